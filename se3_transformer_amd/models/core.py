"""Equivariant primitive modules: Linear / Norm / Conv (TFN) / FeedForward.

Module-tree (state_dict) parity with the reference
(se3_transformer_pytorch.py:67-383) — parameter names, shapes and init
distributions match. The COMPUTE is restructured MI355X-first:

* ConvSE3 never materializes the per-edge (2do+1)*mo x (2di+1)*mi kernel
  matrix (reference :326-343): it contracts basis x features first
  (u = B.x, no channel^2 term) and then the radial output against u,
  streamed over edge chunks. This removes the `splits` machinery
  (reference :190,221-252) entirely — `splits` is accepted and ignored.
* All ops are functional (no in-place masked_fill_).
* On CUDA(ROCm) devices the hot contractions dispatch to fused HIP kernels
  (se3_transformer_amd/ops/fused.py) when the extension is available.
"""
from __future__ import annotations

import os
from math import sqrt

import torch
import torch.nn.functional as F
from torch import nn

from ..utils import (batched_index_select, fourier_encode, masked_mean,
                     rand_uniform, to_order)
from .fiber import Fiber

__all__ = [
    'ResidualSE3', 'LinearSE3', 'NormSE3', 'RadialFunc', 'PairwiseConv',
    'ConvSE3', 'FeedForwardSE3', 'FeedForwardBlockSE3', 'HtypesNorm',
]


def basis_entry(basis, d_in: int, d_out: int):
    """Fetch the (d_in, d_out) basis as a packed (..., 2do+1, 2di+1, F) tensor,
    accepting either the packed dict (tuple keys) or the reference-shaped
    dict ('di,do' keys with broadcast singleton dims)."""
    if (d_in, d_out) in basis:
        return basis[(d_in, d_out)]
    v = basis[f'{d_in},{d_out}']
    # (b,n,k,1,2do+1,1,2di+1,F) -> (b,n,k,2do+1,2di+1,F)
    return v.squeeze(-3).squeeze(-5)


class ResidualSE3(nn.Module):
    """Per-degree residual add (both fibers identical)."""

    def forward(self, x, res):
        out = {}
        for degree, tensor in x.items():
            degree = str(degree)
            out[degree] = tensor + res[degree] if degree in res else tensor
        return out


class LinearSE3(nn.Module):
    """Per-degree channel mixing; weights.<degree> [dim_in, dim_out]."""

    def __init__(self, fiber_in: Fiber, fiber_out: Fiber):
        super().__init__()
        self.weights = nn.ParameterDict()
        for degree, dim_in, dim_out in (fiber_in & fiber_out):
            self.weights[str(degree)] = nn.Parameter(torch.randn(dim_in, dim_out) / sqrt(dim_in))

    def forward(self, x):
        out = {}
        for degree, weight in self.weights.items():
            t = x[degree]
            # (b, n, d, m) x (d, e) -> (b, n, e, m)
            out[degree] = torch.einsum('...dm,de->...em', t, weight.to(t.dtype))
        return out


class NormSE3(nn.Module):
    """Equivariant norm-gated nonlinearity: out = nonlin(scale * |x|) * x/|x|.

    transform.<degree>.{scale|w_gate} parameter names as the reference
    (:123-127); only the parameter actually used is registered.
    """

    def __init__(self, fiber: Fiber, nonlin=nn.GELU(), gated_scale=False, eps=1e-12):
        super().__init__()
        self.fiber = fiber
        self.nonlin = nonlin
        self.eps = eps
        self.gated_scale = gated_scale

        self.transform = nn.ModuleDict()
        for degree, chan in fiber:
            entries = {}
            if not gated_scale:
                entries['scale'] = nn.Parameter(torch.ones(1, 1, chan))
            else:
                entries['w_gate'] = nn.Parameter(rand_uniform((chan, chan), -1e-3, 1e-3))
            self.transform[str(degree)] = nn.ParameterDict(entries)

    def forward(self, features):
        from ..ops.fused import ext_available, norm_se3
        use_fused = (not self.gated_scale and isinstance(self.nonlin, nn.GELU)
                     and getattr(self.nonlin, 'approximate', 'none') == 'none'
                     and os.environ.get('SE3_EAGER_NORM') != '1')
        output = {}
        for degree, t in features.items():
            if (use_fused and t.is_cuda and t.shape[-1] in (1, 3, 5, 7)
                    and t.dtype in (torch.float32, torch.bfloat16)
                    and ext_available()):
                output[degree] = norm_se3(t.contiguous(),
                                          self.transform[degree]['scale'],
                                          self.eps)
                continue
            norm = t.norm(dim=-1, keepdim=True).clamp(min=self.eps)
            phase = t / norm
            params = self.transform[degree]
            if 'scale' in params:
                scale = params['scale'].to(t.dtype)
                transformed = norm.squeeze(-1) * scale
            else:
                w_gate = params['w_gate'].to(t.dtype)
                transformed = torch.einsum('...d,de->...e', norm.squeeze(-1), w_gate)
            transformed = self.nonlin(transformed).unsqueeze(-1)
            output[degree] = (transformed * phase).view(*t.shape)
        return output


class RadialFunc(nn.Module):
    """Per degree-pair radial hypernetwork: edge scalars -> per-frequency,
    per-(channel_in, channel_out) weights. Layer layout (net.0/1/3/4/6) and
    shapes as reference :287-295."""

    def __init__(self, num_freq, in_dim, out_dim, edge_dim=None, mid_dim=128):
        super().__init__()
        self.num_freq = num_freq
        self.in_dim = in_dim
        self.mid_dim = mid_dim
        self.out_dim = out_dim
        self.edge_dim = edge_dim if edge_dim is not None else 0

        self.net = nn.Sequential(
            nn.Linear(self.edge_dim + 1, mid_dim),
            nn.LayerNorm(mid_dim),
            nn.GELU(),
            nn.Linear(mid_dim, mid_dim),
            nn.LayerNorm(mid_dim),
            nn.GELU(),
            nn.Linear(mid_dim, num_freq * in_dim * out_dim),
        )

    def forward(self, x):
        """Returns (..., out_dim, in_dim, num_freq) — o-major, then i, then f,
        matching the reference's '(o i f)' flattening (:297-299) without the
        broadcast singletons."""
        y = self.net(x)
        return y.view(*y.shape[:-1], self.out_dim, self.in_dim, self.num_freq)

    def hidden(self, x):
        """The 128-dim trunk activations (everything but the final Linear) —
        used by the fused HIP path, which folds net.6 into the conv kernel.

        On CUDA(ROCm) the whole trunk (net.0 Linear + net.1 LN + GELU +
        net.3 Linear + net.4 LN + GELU, reference :287-295) runs as ONE
        HIP kernel each way (csrc/radial.hip); eager fallback otherwise."""
        from ..ops import fused as _fused
        in_dim = x.shape[-1]
        if (x.is_cuda and os.environ.get('SE3_EAGER_RADIAL') != '1'
                and _fused.radial_trunk_ok(in_dim, self.mid_dim)):
            n0, ln1 = self.net[0], self.net[1]
            n3, ln4 = self.net[3], self.net[4]
            lead = x.shape[:-1]
            h = _fused.radial_trunk(x.reshape(-1, in_dim),
                                    n0.weight, n0.bias, ln1.weight, ln1.bias,
                                    n3.weight, n3.bias, ln4.weight, ln4.bias,
                                    ln1.eps)
            return h.view(*lead, self.mid_dim)
        h = x
        for layer in self.net[:-1]:
            h = layer(h)
        return h


class PairwiseConv(nn.Module):
    """SE(3)-equivariant per-edge kernel between two single degrees.

    Holds the radial net under `.rp` (state-dict parity, reference :301-343).
    `forward` keeps the reference contract (returns the materialized per-edge
    kernel matrix) for API compatibility, but the framework's compute path
    uses `apply_fused`, which never builds that matrix.
    """

    def __init__(self, degree_in, nc_in, degree_out, nc_out, edge_dim=0, splits=4):
        super().__init__()
        self.degree_in = degree_in
        self.degree_out = degree_out
        self.nc_in = nc_in
        self.nc_out = nc_out
        self.num_freq = to_order(min(degree_in, degree_out))
        self.d_out = to_order(degree_out)
        self.edge_dim = edge_dim
        self.splits = splits  # accepted for API parity; unused
        self.rp = RadialFunc(self.num_freq, nc_in, nc_out, edge_dim)

    def forward(self, feat, basis):
        """Reference-compatible: (..., (2do+1)*nc_out, (2di+1)*nc_in)."""
        r = self.rp(feat)  # (..., mo, mi, F)
        b = basis_entry(basis, self.degree_in, self.degree_out)  # (..., O, I, F)
        k = torch.einsum('...mcf,...oif->...moci', r, b.to(r.dtype))
        return k.reshape(*k.shape[:-4], self.d_out * self.nc_out,
                         to_order(self.degree_in) * self.nc_in)

    def apply_fused(self, edge_feats, basis, x_gathered, max_chunk_bytes=2 << 30):
        """out[..., mo, 2do+1] = sum_{mi,i,f} R[...,mo,mi,f] B[...,o,i,f] x[...,mi,i]

        Structure (mirrors the fused HIP kernel):
          1. u = B.x once per pair — no channel^2 term, small (E*mi*F*O);
          2. the radial output R (the E*mo*mi*F memory hog) is produced and
             consumed chunk-by-chunk and, under grad, recomputed in backward
             (activation checkpointing) instead of being stored.
        """
        O = self.d_out
        I = to_order(self.degree_in)
        F_ = self.num_freq
        mo, mi = self.nc_out, self.nc_in

        b = basis_entry(basis, self.degree_in, self.degree_out)
        lead = x_gathered.shape[:-2]
        e_total = 1
        for s in lead:
            e_total *= s

        ef = edge_feats.reshape(e_total, edge_feats.shape[-1])
        braw = b.reshape(e_total, O, I, F_)
        xg = x_gathered.reshape(e_total, mi, I)

        # MI355X fused HIP path: bf16 compute on CUDA(ROCm) devices
        from ..ops import fused as _fused
        want_bf16 = (torch.is_autocast_enabled()
                     or xg.dtype == torch.bfloat16
                     or os.environ.get('SE3_FORCE_FUSED') == '1')
        if (xg.is_cuda and want_bf16
                and _fused.fused_shapes_ok(mo, mi * F_, O, self.rp.mid_dim)):
            _fused.require_ext()
            h = self.rp.hidden(ef)                           # (E, 128)
            # u_t[(mi,f), o, e] = sum_i B[e,o,i,f] x[e,mi,i]
            if (_fused.ubuild_ok(braw, mi, O, I, F_)
                    and os.environ.get('SE3_EAGER_UBUILD') != '1'):
                u_t = _fused.ubuild(xg, braw.contiguous(), O, I, F_)
            else:
                u_t = torch.einsum('eoif,eci->cfoe',
                                   braw.to(xg.dtype), xg) \
                    .reshape(mi * F_, O, e_total)
            w6 = self.rp.net[6]
            out = _fused.fused_pairconv(h, w6.weight, w6.bias, u_t, mo)
            return out.view(*lead, mo, O).to(xg.dtype)

        # u: (E, mi*F, O), (mi-major, f-minor) matching R's (mo, mi*F) layout
        u = torch.einsum('eoif,eci->ecfo', braw.to(xg.dtype), xg) \
            .reshape(e_total, mi * F_, O)

        elem_size = xg.element_size()
        per_edge = mo * mi * F_ * elem_size
        chunk = max(256, min(e_total, max_chunk_bytes // max(per_edge, 1)))

        def run_chunk(ef_c, u_c):
            r = self.rp(ef_c).to(u_c.dtype).reshape(-1, mo, mi * F_)
            return torch.bmm(r, u_c)                             # (E, mo, O)

        use_ckpt = torch.is_grad_enabled() and (e_total * per_edge > (1 << 26))

        outs = []
        for s in range(0, e_total, chunk):
            e = slice(s, s + chunk)
            if use_ckpt:
                out = torch.utils.checkpoint.checkpoint(
                    run_chunk, ef[e], u[e], use_reentrant=False)
            else:
                out = run_chunk(ef[e], u[e])
            outs.append(out)
        out = torch.cat(outs, dim=0) if len(outs) > 1 else outs[0]
        return out.view(*lead, mo, O)


class ConvSE3(nn.Module):
    """Tensor-field-network convolution over neighbor edges (reference :154-268).

    kernel_unary.'(di,do)' module names, self_interact / self_interact_sum
    naming and semantics match the reference; compute is the fused streaming
    path (see module docstring).
    """

    def __init__(self, fiber_in, fiber_out, self_interaction=True, pool=True,
                 edge_dim=0, fourier_encode_dist=False, num_fourier_features=4,
                 splits=4):
        super().__init__()
        edge_dim = edge_dim if edge_dim is not None else 0
        self.fiber_in = fiber_in
        self.fiber_out = fiber_out
        self.edge_dim = edge_dim
        self.self_interaction = self_interaction
        self.num_fourier_features = num_fourier_features
        self.fourier_encode_dist = fourier_encode_dist
        edge_dim += 0 if not fourier_encode_dist else (num_fourier_features * 2)
        self.kernel_unary = nn.ModuleDict()
        self.splits = splits
        for (di, mi), (do, mo) in (self.fiber_in * self.fiber_out):
            self.kernel_unary[f'({di},{do})'] = PairwiseConv(di, mi, do, mo, edge_dim=edge_dim, splits=splits)
        self.pool = pool
        if self_interaction:
            assert self.pool, 'must pool edges if followed with self interaction'
            self.self_interact = LinearSE3(fiber_in, fiber_out)
            self.self_interact_sum = ResidualSE3()

    _stream_pool = {}

    @staticmethod
    def _streams(device, k=4):
        pool = ConvSE3._stream_pool.get((device, k))
        if pool is None:
            pool = [torch.cuda.Stream(device=device) for _ in range(k)]
            ConvSE3._stream_pool[(device, k)] = pool
        return pool

    def forward(self, inp, edge_info, rel_dist=None, basis=None):
        neighbor_indices, neighbor_masks, edges = edge_info
        rel = rel_dist.unsqueeze(-1)
        if self.fourier_encode_dist:
            rel = fourier_encode(rel[..., None], num_encodings=self.num_fourier_features)
        edge_feats = torch.cat((rel, edges), dim=-1) if edges is not None else rel

        # gather neighbor features once per input degree
        gathered = {}
        for di, _ in self.fiber_in:
            x = batched_index_select(inp[str(di)], neighbor_indices, dim=1)
            gathered[di] = x  # (b, n, k, mi, 2di+1)

        pairs = list(self.fiber_in * self.fiber_out)
        # opt-in: overlaps independent pairs on a stream pool, but is NOT
        # compatible with hipGraph step capture (capture with forked streams
        # hangs on ROCm 7.2) — default off
        use_streams = (edge_feats.is_cuda and len(pairs) > 1
                       and os.environ.get('SE3_STREAMS') == '1'
                       # forked streams hang under hipGraph capture on ROCm
                       # 7.2: fall back to the single-stream loop mid-capture
                       and not torch.cuda.is_current_stream_capturing())

        pair_out = {}
        if use_streams:
            # the degree pairs are data-independent: fan them out over a small
            # HIP stream pool so one pair's stalls overlap another's compute
            cur = torch.cuda.current_stream()
            pool = self._streams(edge_feats.device)
            for idx, ((di, _mi), (do, _mo)) in enumerate(pairs):
                s = pool[idx % len(pool)]
                s.wait_stream(cur)
                with torch.cuda.stream(s):
                    pc = self.kernel_unary[f'({di},{do})']
                    pair_out[(di, do)] = pc.apply_fused(edge_feats, basis, gathered[di])
            for s in pool:
                cur.wait_stream(s)
            for t in pair_out.values():
                t.record_stream(cur)
        else:
            for (di, _mi), (do, _mo) in pairs:
                pc = self.kernel_unary[f'({di},{do})']
                pair_out[(di, do)] = pc.apply_fused(edge_feats, basis, gathered[di])

        outputs = {}
        for do, _mo in self.fiber_out:
            acc = None
            for di, _mi in self.fiber_in:
                out = pair_out[(di, do)]
                acc = out if acc is None else acc + out

            if self.pool:
                acc = masked_mean(acc, neighbor_masks, dim=2) if neighbor_masks is not None else acc.mean(dim=2)
            outputs[str(do)] = acc

        if self.self_interaction:
            outputs = self.self_interact_sum(outputs, self.self_interact(inp))
        return outputs


class FeedForwardSE3(nn.Module):
    def __init__(self, fiber, mult=4):
        super().__init__()
        self.fiber = fiber
        fiber_hidden = Fiber([(degree, dim * mult) for degree, dim in fiber])
        self.project_in = LinearSE3(fiber, fiber_hidden)
        self.nonlin = NormSE3(fiber_hidden)
        self.project_out = LinearSE3(fiber_hidden, fiber)

    def forward(self, features):
        return self.project_out(self.nonlin(self.project_in(features)))


class FeedForwardBlockSE3(nn.Module):
    def __init__(self, fiber, norm_gated_scale=False):
        super().__init__()
        self.fiber = fiber
        self.prenorm = NormSE3(fiber, gated_scale=norm_gated_scale)
        self.feedforward = FeedForwardSE3(fiber)
        self.residual = ResidualSE3()

    def forward(self, features):
        res = features
        out = self.prenorm(features)
        out = self.feedforward(out)
        return self.residual(out, res)


class HtypesNorm(nn.Module):
    """Norm/renorm of higher-type vectors with learned scale + bias
    (reference :693-705)."""

    def __init__(self, dim, eps=1e-8, scale_init=1e-2, bias_init=1e-2):
        super().__init__()
        self.eps = eps
        self.scale = nn.Parameter(torch.full((1, 1, 1, dim, 1), scale_init))
        self.bias = nn.Parameter(torch.full((1, 1, 1, dim, 1), bias_init))

    def forward(self, coors):
        norm = coors.norm(dim=-1, keepdim=True)
        normed = coors / norm.clamp(min=self.eps)
        return normed * (norm * self.scale + self.bias)
