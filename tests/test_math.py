"""Math-layer property tests: spherical harmonics, Wigner-D, Q_J intertwiners.

Mirrors the reference test strategy (tests/test_irrep_repr.py,
tests/test_basis.py, tests/test_spherical_harmonics.py) plus extra
self-consistency oracles (representation property, scipy cross-check).
"""
import math

import pytest
import torch

from se3_transformer_amd.ops import (
    basis_transformation_Q_J, compose, get_basis, get_basis_packed,
    get_R_tensor, get_spherical_harmonics, get_spherical_harmonics_element,
    irr_repr, rot, spherical_harmonics, wigner_d,
)
from se3_transformer_amd.ops.sh import sh_packed_from_cartesian, sh_offset
from se3_transformer_amd.utils import torch_default_dtype, to_order


@torch_default_dtype(torch.float64)
def test_sh_against_scipy():
    """Tesseral SH must match scipy's sph_harm-derived real harmonics."""
    from scipy.special import sph_harm
    theta = torch.rand(64, dtype=torch.float64) * math.pi
    phi = (torch.rand(64, dtype=torch.float64) * 2 - 1) * math.pi
    for l in range(8):
        for m in range(-l, l + 1):
            y = get_spherical_harmonics_element(l, m, theta, phi).numpy()
            # scipy: sph_harm(m, l, azimuth, polar) complex with CS phase
            z = sph_harm(abs(m), l, phi.numpy(), theta.numpy())
            # NB: the reference convention keeps the CS phase inside P_l^m and
            # does NOT apply the extra (-1)^m of the textbook real harmonics:
            # Y_{l,m>0} = sqrt2*Re(Y_complex), Y_{l,m<0} = sqrt2*Im(Y_complex).
            if m > 0:
                ref = math.sqrt(2) * z.real
            elif m < 0:
                ref = math.sqrt(2) * z.imag
            else:
                ref = z.real
            err = abs(y - ref).max()
            assert err < 1e-10, (l, m, err)


@torch_default_dtype(torch.float64)
def test_sh_cartesian_matches_angle_path():
    """Algebraic cartesian SH == angle-based SH through the reference's
    axis permutation + theta = pi - beta chain."""
    r = torch.randn(256, 3, dtype=torch.float64)
    x_sh, y_sh, z_sh = r[:, 2], r[:, 0], r[:, 1]
    beta = torch.atan2(torch.sqrt(x_sh ** 2 + y_sh ** 2), z_sh)
    alpha = torch.atan2(y_sh, x_sh)
    packed = sh_packed_from_cartesian(6, r)
    for l in range(7):
        ref = get_spherical_harmonics(l, math.pi - beta, alpha)
        got = packed[:, sh_offset(l): sh_offset(l + 1)]
        assert (ref - got).abs().max() < 1e-10, l


@torch_default_dtype(torch.float64)
def test_degree1_sh_is_negative_identity():
    """Y_1(v) must be proportional to -v in STANDARD (x,y,z) order — this is
    what makes D^1(R) = R and the `out @ R` equivariance contract work."""
    v = torch.randn(32, 3, dtype=torch.float64)
    v = v / v.norm(dim=-1, keepdim=True)
    y1 = sh_packed_from_cartesian(1, v)[:, 1:4]
    n = math.sqrt(3 / (4 * math.pi))
    assert (y1 + n * v).abs().max() < 1e-12


@torch_default_dtype(torch.float64)
def test_irr_repr_compose_property():
    """Y(Z(a) Y(b) Z(c) x) = D(a,b,c) Y(x) — reference tests/test_irrep_repr.py."""
    g = torch.Generator().manual_seed(7)
    for order in range(7):
        a, b = torch.rand(2, generator=g, dtype=torch.float64)
        alpha, beta, gamma = torch.rand(3, generator=g, dtype=torch.float64)
        ra, rb, _ = compose(alpha, beta, gamma, a, b, 0)
        y_rx = spherical_harmonics(order, ra, rb)
        y = spherical_harmonics(order, a, b)
        d_r_y = irr_repr(order, alpha, beta, gamma) @ y
        d, r = (y_rx - d_r_y).abs().max(), y.abs().max()
        assert d < 1e-10 * r, (order, (d / r).item())


@torch_default_dtype(torch.float64)
def test_wigner_is_representation():
    """D(R1)D(R2) = D(R1 R2) and orthogonality."""
    g = torch.Generator().manual_seed(3)
    a1, b1, c1, a2, b2, c2 = torch.rand(6, generator=g, dtype=torch.float64) * 4
    for order in range(5):
        d1 = wigner_d(order, a1, b1, c1)
        d2 = wigner_d(order, a2, b2, c2)
        ab, bb, cb = compose(a1, b1, c1, a2, b2, c2)
        d12 = wigner_d(order, ab, bb, cb)
        assert (d1 @ d2 - d12).abs().max() < 1e-9
        eye = torch.eye(to_order(order), dtype=torch.float64)
        assert (d1 @ d1.t() - eye).abs().max() < 1e-10


@torch_default_dtype(torch.float64)
def test_qj_intertwiner_property():
    """(D_out ⊗ D_in) Q_J = Q_J D_J at random angles — reference tests/test_basis.py:11.
    (get_R_tensor follows the default dtype, as the reference's does.)"""
    g = torch.Generator().manual_seed(11)
    rand_angles = torch.rand(4, 3, generator=g, dtype=torch.float64)
    for (J, order_in, order_out) in [(1, 1, 1), (2, 1, 1), (3, 2, 1), (2, 2, 2)]:
        q_j = basis_transformation_Q_J(J, order_in, order_out).double()
        for a, b, c in rand_angles:
            lhs = get_R_tensor(order_out, order_in, a, b, c) @ q_j
            rhs = q_j @ irr_repr(J, a, b, c, dtype=torch.float64)
            assert torch.allclose(lhs, rhs, atol=1e-8), (J, order_in, order_out)


def test_basis_dict_shape():
    max_degree = 3
    x = torch.randn(2, 8, 4, 3)
    basis = get_basis(x, max_degree)
    assert len(basis) == (max_degree + 1) ** 2
    for d_in in range(max_degree + 1):
        for d_out in range(max_degree + 1):
            v = basis[f'{d_in},{d_out}']
            f = 2 * min(d_in, d_out) + 1
            assert v.shape == (2, 8, 4, 1, 2 * d_out + 1, 1, 2 * d_in + 1, f)


def test_basis_packed_matches_dict():
    x = torch.randn(2, 6, 3, 3)
    b1 = get_basis(x, 2)
    b2 = get_basis_packed(x, 2)
    for (d_in, d_out), v in b2.items():
        ref = b1[f'{d_in},{d_out}'].squeeze(-2).squeeze(-4)
        # ref view: (b,n,k, 2do+1, 2di+1, F)
        assert torch.allclose(ref.reshape(v.shape), v, atol=1e-6)


def test_basis_differentiable_flag():
    x = torch.randn(2, 4, 3, 3, requires_grad=True)
    basis = get_basis(x, 1, differentiable=True)
    loss = sum(v.sum() for v in basis.values())
    loss.backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()
    # non-differentiable path must be detached
    x2 = torch.randn(2, 4, 3, 3, requires_grad=True)
    basis2 = get_basis(x2, 1, differentiable=False)
    assert all(not v.requires_grad for v in basis2.values())


@torch_default_dtype(torch.float64)
def test_basis_kernel_equivariance():
    """The assembled kernel must satisfy K(R r) = D_out K(r) D_in^T component-wise
    in each frequency channel's learned-combination sense: check via the full
    contraction with random radial weights."""
    torch.manual_seed(5)
    r = torch.randn(16, 3, dtype=torch.float64)
    angles = torch.rand(3, dtype=torch.float64) * 4
    R3 = rot(*angles)
    d_in, d_out = 1, 2
    f = 2 * min(d_in, d_out) + 1
    w = torch.randn(f, dtype=torch.float64)

    def kernel(rr):
        b = get_basis_packed(rr, 2)[(d_in, d_out)].double()  # [n, 2do+1, 2di+1, F]
        return (b * w).sum(-1)

    k1 = kernel(r @ R3)  # rotated rel positions (row convention as the model)
    k0 = kernel(r)
    # row-vector rotation x' = x @ R is column rotation M = R^T; the basis SH
    # are evaluated in the permuted frame (sh x,y,z = std z,x,y), so the
    # covariance matrices are D(P M P^-1) with P the permutation.
    from se3_transformer_amd.ops.wigner import wigner_d_from_matrix
    P = torch.tensor([[0., 0., 1.], [1., 0., 0.], [0., 1., 0.]], dtype=torch.float64)
    m_perm = P @ R3.t() @ P.t()
    d_o = wigner_d_from_matrix(d_out, m_perm)
    d_i = wigner_d_from_matrix(d_in, m_perm)
    rhs = torch.einsum('oi,nij,pj->nop', d_o, k0, d_i)
    assert (k1 - rhs).abs().max() < 1e-8


def test_utils_fast_split_and_fourier():
    """Reference utils parity: fast_split chunking (utils.py:85-94) and
    fourier_encode (utils.py:96-104)."""
    import torch
    from se3_transformer_amd.utils import fast_split, fourier_encode

    t = torch.arange(10).float().unsqueeze(-1)
    chunks = list(fast_split(t, 4, dim=0))
    assert sum(c.shape[0] for c in chunks) == 10
    assert torch.equal(torch.cat(chunks, dim=0), t)

    # reference flatten keeps the first three dims: 'b m n ... -> b m n (...)'
    x = torch.randn(3, 5, 2, 4)
    enc = fourier_encode(x, num_encodings=4, include_self=True)
    assert enc.shape == (3, 5, 2, 4 * (2 * 4 + 1))
    enc2 = fourier_encode(torch.randn(3, 5), num_encodings=2,
                          include_self=False, flatten=False)
    assert enc2.shape == (3, 5, 4)


def test_utils_batched_index_select_masked_mean():
    import torch
    from se3_transformer_amd.utils import batched_index_select, masked_mean

    v = torch.randn(2, 6, 3, 5)
    idx = torch.randint(0, 6, (2, 4))
    out = batched_index_select(v, idx, dim=1)
    for b in range(2):
        for i in range(4):
            assert torch.equal(out[b, i], v[b, idx[b, i]])

    t = torch.randn(2, 4, 3)
    m = torch.tensor([[True, True, False, False], [True, False, False, False]])
    mm = masked_mean(t, m, dim=1)
    assert torch.allclose(mm[0], t[0, :2].mean(dim=0), atol=1e-6)
    assert torch.allclose(mm[1], t[1, :1].mean(dim=0), atol=1e-6)
    # functional: input not mutated (reference masked_fill_ mutates — SURVEY §2.4)
    assert torch.isfinite(t).all()


def test_basis_cache_survives_inference_mode():
    """A basis computed under torch.inference_mode (e.g. a serving request)
    must not poison the module-level Q_J/table caches for later training:
    the cached device tensors are built with inference mode forced off."""
    from se3_transformer_amd.ops import basis as basis_mod
    basis_mod._qj_dev_cache.clear()
    basis_mod._sh_tables_cache.clear()

    with torch.inference_mode():
        get_basis(torch.randn(2, 3, 4, 3), 2)           # fills the caches
        from se3_transformer_amd.ops.basis import get_basis_packed
        get_basis_packed(torch.randn(2, 3, 4, 3), 2)

    assert all(not t.is_inference() for t in basis_mod._qj_dev_cache.values())

    x = torch.randn(2, 3, 4, 3, requires_grad=True)
    loss = sum(v.sum() for v in get_basis(x, 2, differentiable=True).values())
    loss.backward()                                      # raised before the fix
    assert x.grad is not None and torch.isfinite(x.grad).all()
