"""Equivariant weight basis: Q_J intertwiners and per-edge basis kernels.

Functional contract follows reference /root/reference/se3_transformer_pytorch/basis.py
(get_basis at :153, basis_transformation_Q_J at :123) with these deliberate
changes for the MI355X build:

* Q_J is solved against our own least-squares Wigner-D (ops/wigner.py) — no
  J_dense blobs, no lie_learn; sign fixed deterministically so every DP rank
  computes bit-identical tables on CPU float64.
* ``differentiable=True`` genuinely keeps the autograd graph through the
  spherical harmonics (the reference's flag is inverted/defeated:
  basis.py:171 vs :200-203).
* Spherical harmonics for all J are evaluated in ONE vectorized pass straight
  from cartesian offsets (no per-(l,m) recursion / global cache) — the same
  structure as the fused HIP basis kernel.
* A packed per-pair layout (`get_basis_packed`) is provided for the fused
  conv kernels; `get_basis` returns the reference-shaped dict
  (b, n, k, 1, 2*do+1, 1, 2*di+1, 2*min(di,do)+1).

Disk persistence of Q_J is optional via the CACHE_PATH env var (reference
basis.py:15-16 keeps the same env name); writes are atomic (tmp+rename), no
file locks needed.
"""
from __future__ import annotations

import os
import tempfile
from functools import lru_cache
from itertools import product

import torch

from ..utils import to_order, torch_default_dtype
from .sh import sh_packed_from_cartesian, sh_offset
from .wigner import wigner_d

__all__ = ['basis_transformation_Q_J', 'get_basis', 'get_basis_packed', 'get_R_tensor', 'num_basis_freq']

# deterministic generic Euler angles for the intertwiner solve (any 5 generic
# rotations characterize the intertwiner space; fixed seed => reproducible)
_ANGLE_SEED = 271828


def _solver_angles():
    g = torch.Generator().manual_seed(_ANGLE_SEED)
    return (torch.rand(5, 3, generator=g, dtype=torch.float64) * 6.28).tolist()


def num_basis_freq(d_in: int, d_out: int) -> int:
    return to_order(min(d_in, d_out))


def get_R_tensor(order_out: int, order_in: int, a, b, c) -> torch.Tensor:
    """Kronecker product D_out(R) ⊗ D_in(R) (reference basis.py:110).
    Returned in the caller's default dtype, like the reference (the Q_J
    solver runs under a float64 default-dtype context)."""
    return torch.kron(wigner_d(order_out, a, b, c),
                      wigner_d(order_in, a, b, c)).to(torch.get_default_dtype())


def _null_space_1d(mats, eps=1e-9) -> torch.Tensor:
    """The single common null vector of the stacked matrices (f64 SVD)."""
    a = torch.cat(mats, dim=0)
    _, s, vh = torch.linalg.svd(a)
    # rows of vh beyond rank are the null space; with a 1-D null space the
    # smallest-singular-value row is the solution
    assert s[-1] < eps and (s.shape[0] < 2 or s[-2] > eps), \
        f'intertwiner null space is not 1-dimensional (singular values tail: {s[-3:]})'
    return vh[-1]


def _cache_file():
    path = os.environ.get('CACHE_PATH', os.path.expanduser('~/.cache/se3_transformer_amd'))
    if os.environ.get('CLEAR_CACHE') is not None:
        return None
    return os.path.join(path, 'qj_tables_v1.pt')


_qj_disk = None


def _load_disk_cache():
    global _qj_disk
    if _qj_disk is not None:
        return _qj_disk
    _qj_disk = {}
    f = _cache_file()
    if f is not None and os.path.exists(f):
        try:
            _qj_disk = torch.load(f, weights_only=True)
        except Exception:
            _qj_disk = {}
    return _qj_disk


def _save_disk_cache():
    f = _cache_file()
    if f is None:
        return
    try:
        os.makedirs(os.path.dirname(f), exist_ok=True)
        fd, tmp = tempfile.mkstemp(dir=os.path.dirname(f))
        os.close(fd)
        torch.save(_qj_disk, tmp)
        os.replace(tmp, f)
    except OSError:
        pass


@lru_cache(maxsize=None)
def basis_transformation_Q_J(J: int, order_in: int, order_out: int) -> torch.Tensor:
    """Q_J: [ (2*order_out+1)*(2*order_in+1), 2J+1 ] float32 intertwiner with
    (D_out ⊗ D_in) Q_J = Q_J D_J for every rotation (reference basis.py:123).
    """
    key = (J, order_in, order_out)
    disk = _load_disk_cache()
    if key in disk:
        return disk[key]

    with torch_default_dtype(torch.float64), torch.no_grad():
        mats = []
        for a, b, c in _solver_angles():
            r_tensor = get_R_tensor(order_out, order_in, a, b, c)
            d_j = wigner_d(J, a, b, c)
            eye_r = torch.eye(r_tensor.shape[0], dtype=torch.float64)
            eye_j = torch.eye(d_j.shape[0], dtype=torch.float64)
            # vec(X): (R ⊗ I) vec - (I ⊗ D_J^T) vec = 0  <=>  R X = X D_J
            mats.append(torch.kron(r_tensor.contiguous(), eye_j)
                        - torch.kron(eye_r, d_j.t().contiguous()))
        q = _null_space_1d(mats)
        # deterministic sign: largest-|entry| component positive
        idx = q.abs().argmax()
        if q[idx] < 0:
            q = -q
        q = q.view(to_order(order_out) * to_order(order_in), to_order(J)).float().contiguous()

    disk[key] = q
    _save_disk_cache()
    return q


@lru_cache(maxsize=None)
def _qj_transposed(J: int, d_in: int, d_out: int) -> torch.Tensor:
    return basis_transformation_Q_J(J, d_in, d_out).t().contiguous()


_qj_dev_cache = {}


def _canon_device(device) -> str:
    """Canonical device key: 'cuda' and 'cuda:0' must not duplicate cache
    entries."""
    d = torch.device(device)
    if d.type == 'cuda' and d.index is None:
        d = torch.device('cuda', torch.cuda.current_device())
    return str(d)


def _qj_t_on(J: int, d_in: int, d_out: int, device, dtype) -> torch.Tensor:
    """Device/dtype-resident Q_J^T (cached so steady-state forwards — and
    hipGraph capture — never issue host-to-device copies)."""
    key = (J, d_in, d_out, _canon_device(device), dtype)
    t = _qj_dev_cache.get(key)
    if t is None:
        # fill outside inference mode: a cache populated during a
        # torch.inference_mode() forward (e.g. a serving request) would
        # otherwise hold inference tensors that poison any later training
        # step in the same process ("Inference tensors cannot be saved
        # for backward")
        with torch.inference_mode(False):
            t = _qj_transposed(J, d_in, d_out).to(device=device, dtype=dtype)
        _qj_dev_cache[key] = t
    return t


def _compute_sh(r_ij: torch.Tensor, max_J: int, differentiable: bool) -> torch.Tensor:
    if differentiable:
        return sh_packed_from_cartesian(max_J, r_ij)
    with torch.no_grad():
        return sh_packed_from_cartesian(max_J, r_ij)


def get_basis(r_ij: torch.Tensor, max_degree: int, differentiable: bool = False):
    """Reference-shaped equivariant basis dict.

    Keys '{d_in},{d_out}' for d_in, d_out in 0..max_degree; values of shape
    (*r_ij.shape[:-1], 1, 2*d_out+1, 1, 2*d_in+1, 2*min(d_in,d_out)+1)
    (reference basis.py:153-205).
    """
    device, dtype = r_ij.device, r_ij.dtype
    y_packed = _compute_sh(r_ij, 2 * max_degree, differentiable)

    basis = {}
    for d_in, d_out in product(range(max_degree + 1), range(max_degree + 1)):
        k_js = []
        for J in range(abs(d_in - d_out), d_in + d_out + 1):
            q_t = _qj_t_on(J, d_in, d_out, device, dtype)
            y_j = y_packed[..., sh_offset(J): sh_offset(J + 1)]
            k_js.append(y_j @ q_t)  # [..., (2do+1)(2di+1)]
        k = torch.stack(k_js, dim=-1)  # [..., (2do+1)(2di+1), F]
        size = (*r_ij.shape[:-1], 1, to_order(d_out), 1, to_order(d_in),
                num_basis_freq(d_in, d_out))
        basis[f'{d_in},{d_out}'] = k.view(*size)

    if not differentiable:
        basis = {k: v.detach() for k, v in basis.items()}
    return basis


_sh_tables_cache = {}


def _sh_basis_tables(max_degree: int, device):
    """Device-resident Q_J^T concat + per-pair meta + SH norm table for the
    fused sh_basis HIP kernel (cached; hipGraph-capture safe)."""
    key = (max_degree, _canon_device(device))
    hit = _sh_tables_cache.get(key)
    if hit is not None:
        return hit
    return _build_sh_tables(key, max_degree, device)


@torch.inference_mode(False)
def _build_sh_tables(key, max_degree, device):
    # built outside inference mode so the cached tables stay usable by
    # training steps after a torch.inference_mode() forward (see _qj_t_on)
    L = 2 * max_degree
    qparts, meta, layout = [], [], {}
    off_out = off_q = 0
    for d_in, d_out in product(range(max_degree + 1), range(max_degree + 1)):
        O, I = to_order(d_out), to_order(d_in)
        F = num_basis_freq(d_in, d_out)
        meta.append((d_in, d_out, off_out, off_q))
        layout[(d_in, d_out)] = (off_out, O, I, F)
        for J in range(abs(d_in - d_out), d_in + d_out + 1):
            q_t = _qj_transposed(J, d_in, d_out).float()   # (2J+1, O*I)
            qparts.append(q_t.reshape(-1))
            off_q += q_t.numel()
        off_out += O * I * F
    from .sh import _norm_const
    normtab = torch.tensor([_norm_const(l, m) for l in range(L + 1)
                            for m in range(l + 1)], dtype=torch.float32)
    tables = (torch.cat(qparts).to(device),
              normtab.to(device),
              torch.tensor(meta, dtype=torch.int32, device=device),
              off_out, layout)
    _sh_tables_cache[key] = tables
    return tables


def get_basis_packed(r_ij: torch.Tensor, max_degree: int, differentiable: bool = False):
    """Compact per-pair basis for the fused conv path.

    Returns {(d_in, d_out): tensor [..., 2*d_out+1, 2*d_in+1, F]} with
    F = 2*min(d_in,d_out)+1, contiguous, no broadcast singleton dims.

    On CUDA fp32 non-differentiable inputs this runs the fused HIP
    spherical-harmonics + basis kernel (csrc/sh_basis.hip) in one launch.
    """
    device, dtype = r_ij.device, r_ij.dtype

    if (r_ij.is_cuda and not differentiable and dtype == torch.float32
            and os.environ.get('SE3_EAGER_BASIS') != '1'):
        from . import fused as _fused
        if _fused.ext_available():
            qcat, normtab, meta, TOT, layout = _sh_basis_tables(max_degree, device)
            rel = r_ij.reshape(-1, 3).contiguous()
            E = rel.shape[0]
            out = torch.empty(E, TOT, dtype=torch.float32, device=device)
            _fused._EXT.sh_basis_fwd(rel, qcat, normtab, meta, out, 2 * max_degree)
            basis = {}
            for (d_in, d_out), (off, O, I, F) in layout.items():
                basis[(d_in, d_out)] = out[:, off:off + O * I * F] \
                    .reshape(*r_ij.shape[:-1], O, I, F)
            return basis
    y_packed = _compute_sh(r_ij, 2 * max_degree, differentiable)

    basis = {}
    for d_in, d_out in product(range(max_degree + 1), range(max_degree + 1)):
        k_js = []
        for J in range(abs(d_in - d_out), d_in + d_out + 1):
            q_t = _qj_t_on(J, d_in, d_out, device, dtype)
            y_j = y_packed[..., sh_offset(J): sh_offset(J + 1)]
            k_js.append(y_j @ q_t)
        k = torch.stack(k_js, dim=-1)
        k = k.view(*r_ij.shape[:-1], to_order(d_out), to_order(d_in),
                   num_basis_freq(d_in, d_out))
        basis[(d_in, d_out)] = k if differentiable else k.detach()
    return basis
