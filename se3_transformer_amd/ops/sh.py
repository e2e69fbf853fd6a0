"""Real (tesseral) spherical harmonics, Condon–Shortley phase.

Convention parity with the reference implementation
(/root/reference/se3_transformer_pytorch/spherical_harmonics.py:75-123 and
basis.py:57-95):

* angle API: ``Y_l^m(theta, phi)`` with theta the colatitude, phi the azimuth,
  normalization ``sqrt((2l+1)/(4pi) * (l-|m|)!/(l+|m|)!) * sqrt(2 for m!=0)``,
  cos(m phi) for m>0 and sin(|m| phi) for m<0, associated Legendre with
  Condon–Shortley phase.
* cartesian API: the basis path maps a standard (x,y,z) relative position
  through the reference's axis permutation (SH-frame x,y,z = std z,x,y;
  basis.py:76) and evaluates at theta = pi - beta, phi = alpha
  (irr_repr.py:103-104). With that convention Y_1 is exactly
  -N*(x,y,z)/r in STANDARD order, hence D^1(R) = R — which is what makes the
  ``out @ R`` equivariance contract of the reference tests hold.

Unlike the reference there is NO module-global cache keyed by (l,m)
(spherical_harmonics.py:11-34) — everything is computed in a single
vectorized pass, which is also the structure of the fused HIP kernel.
All functions are differentiable.
"""
from __future__ import annotations

from functools import lru_cache
from math import pi, sqrt

import torch

__all__ = [
    'sh_packed_from_angles', 'sh_packed_from_cartesian', 'sh_list_from_packed',
    'get_spherical_harmonics', 'get_spherical_harmonics_element',
    'clear_spherical_harmonics_cache', 'sh_offset',
]


@lru_cache(maxsize=None)
def _semifactorial(x: int) -> float:
    out = 1.
    while x > 1:
        out *= x
        x -= 2
    return out


@lru_cache(maxsize=None)
def _norm_const(l: int, m: int) -> float:
    """sqrt((2l+1)/(4pi)) and, for m != 0, the extra sqrt(2 (l-m)!/(l+m)!)."""
    m = abs(m)
    n = sqrt((2 * l + 1) / (4 * pi))
    if m == 0:
        return n
    ratio = 1.
    for j in range(l - m + 1, l + m + 1):
        ratio *= j
    return n * sqrt(2. / ratio)


def sh_offset(l: int) -> int:
    """Offset of degree l inside the packed last dim (l-major, m = -l..l)."""
    return l * l


def _legendre_all(L: int, ct: torch.Tensor, st: torch.Tensor):
    """Associated Legendre P_l^m(ct) with CS phase for all 0<=m<=l<=L.

    ``st`` must equal sqrt(1-ct^2) >= 0 (supplied separately so the caller
    can compute it stably from cartesian inputs). Returns {(l, m): tensor}.
    """
    P = {(0, 0): torch.ones_like(ct)}
    # diagonal: P_m^m = (-1)^m (2m-1)!! st^m
    for m in range(1, L + 1):
        P[(m, m)] = P[(m - 1, m - 1)] * st * (-(2 * m - 1))
    # first off-diagonal: P_{m+1}^m = (2m+1) ct P_m^m
    for m in range(0, L):
        P[(m + 1, m)] = (2 * m + 1) * ct * P[(m, m)]
    # upward recursion in l
    for m in range(0, L + 1):
        for l in range(m + 2, L + 1):
            P[(l, m)] = ((2 * l - 1) * ct * P[(l - 1, m)]
                         - (l + m - 1) * P[(l - 2, m)]) / (l - m)
    return P


def _azimuth_all(L: int, cp: torch.Tensor, sp: torch.Tensor):
    """cos(m phi), sin(m phi) for m = 0..L via Chebyshev recurrence on (cp, sp)."""
    cos_m = [torch.ones_like(cp)]
    sin_m = [torch.zeros_like(sp)]
    for m in range(1, L + 1):
        cos_m.append(cos_m[-1] * cp - sin_m[-1] * sp)
        sin_m.append(sin_m[-1] * cp + cos_m[-2] * sp)
    return cos_m, sin_m


def _assemble(L: int, P, cos_m, sin_m, like: torch.Tensor) -> torch.Tensor:
    """Pack Y_l^m into [..., (L+1)^2], l-major, m = -l..l."""
    cols = []
    for l in range(L + 1):
        for m in range(-l, l + 1):
            ma = abs(m)
            leg = P[(l, ma)]
            if m == 0:
                cols.append(_norm_const(l, 0) * leg)
            elif m > 0:
                cols.append(_norm_const(l, m) * leg * cos_m[m])
            else:
                cols.append(_norm_const(l, ma) * leg * sin_m[ma])
    return torch.stack(cols, dim=-1)


def sh_packed_from_angles(L: int, theta: torch.Tensor, phi: torch.Tensor) -> torch.Tensor:
    """All Y_l^m for l <= L at (theta, phi); returns [..., (L+1)^2]."""
    ct = torch.cos(theta)
    st = torch.sin(theta).abs()  # theta in [0, pi] => sin >= 0; abs for safety
    cp, sp = torch.cos(phi), torch.sin(phi)
    P = _legendre_all(L, ct, st)
    cos_m, sin_m = _azimuth_all(L, cp, sp)
    return _assemble(L, P, cos_m, sin_m, theta)


def sh_packed_from_cartesian(L: int, rel_pos: torch.Tensor, eps: float = 1e-12) -> torch.Tensor:
    """All Y_l^m for l <= L of a standard (x,y,z) relative position, algebraic.

    Implements the reference convention chain (basis.py:57-95 axis permutation
    + theta = pi - beta) without any trigonometric calls:
        x_sh, y_sh, z_sh = z, x, y   (standard components)
        cos(theta) = -z_sh / r,  sin(theta) = rho / r,
        cos(phi) = x_sh / rho,   sin(phi) = y_sh / rho.
    Degenerate points (r ~ 0 or rho ~ 0) get cos(phi)=1, sin(phi)=0 and, for
    r ~ 0, cos(theta) = -1 (matching atan2(0,0) = 0 in the reference).
    """
    x_sh = rel_pos[..., 2]
    y_sh = rel_pos[..., 0]
    z_sh = rel_pos[..., 1]

    rho2 = x_sh * x_sh + y_sh * y_sh
    r2 = rho2 + z_sh * z_sh
    rho = torch.sqrt(rho2.clamp(min=eps * eps))
    r = torch.sqrt(r2.clamp(min=eps * eps))

    tiny_rho = rho2 <= (eps * eps)
    tiny_r = r2 <= (eps * eps)

    ct = torch.where(tiny_r, torch.full_like(r, -1.), -z_sh / r)
    st = torch.where(tiny_r, torch.zeros_like(r), rho / r)
    cp = torch.where(tiny_rho, torch.ones_like(rho), x_sh / rho)
    sp = torch.where(tiny_rho, torch.zeros_like(rho), y_sh / rho)

    P = _legendre_all(L, ct, st)
    cos_m, sin_m = _azimuth_all(L, cp, sp)
    return _assemble(L, P, cos_m, sin_m, rel_pos)


def sh_list_from_packed(L: int, packed: torch.Tensor):
    """Split a packed [..., (L+1)^2] tensor into {J: [..., 2J+1]}."""
    return {l: packed[..., sh_offset(l): sh_offset(l + 1)] for l in range(L + 1)}


# ------------------------------------------------------------------
# reference-compatible per-degree API (used by tests and irr_repr)
# ------------------------------------------------------------------

def get_spherical_harmonics(l: int, theta: torch.Tensor, phi: torch.Tensor) -> torch.Tensor:
    """Y_l at (theta, phi): [..., 2l+1] (reference spherical_harmonics.py:108)."""
    return sh_packed_from_angles(l, theta, phi)[..., sh_offset(l):]


def get_spherical_harmonics_element(l: int, m: int, theta: torch.Tensor, phi: torch.Tensor) -> torch.Tensor:
    """Single Y_l^m (reference spherical_harmonics.py:75)."""
    assert abs(m) <= l
    ct = torch.cos(theta)
    st = torch.sin(theta).abs()
    P = _legendre_all(l, ct, st)
    leg = P[(l, abs(m))]
    if m == 0:
        return _norm_const(l, 0) * leg
    if m > 0:
        return _norm_const(l, m) * leg * torch.cos(m * phi)
    return _norm_const(l, m) * leg * torch.sin(abs(m) * phi)


def clear_spherical_harmonics_cache():
    """No-op: this implementation holds no argument-dependent global cache
    (the reference's CACHE keyed only by (l,m) — spherical_harmonics.py:11-17 —
    is a footgun in a multi-stream world and does not exist here)."""
