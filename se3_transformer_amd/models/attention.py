"""Equivariant multi-head attention over k-NN neighborhoods.

Module tree & semantics parity: reference se3_transformer_pytorch.py:387-683
(AttentionSE3 :387, OneHeadedKVAttentionSE3 :522, AttentionBlockSE3 :656).
The per-degree logits/softmax/weighted-sum will dispatch to a fused HIP
neighbor-attention kernel on GPU; the einsum path below is the oracle.
"""
from __future__ import annotations

import os

import torch
import torch.nn.functional as F
from torch import nn

from ..utils import batched_index_select, map_values, to_order
from .core import ConvSE3, LinearSE3, NormSE3, ResidualSE3
from .fiber import Fiber


def apply_rotary_pos_emb(t, freqs):
    """Eager rotary rotation over the leading `rot_dim` feature rows — the
    fallback/oracle for the in-kernel rotation (csrc/attn2.hip rope_apply;
    semantics of reference rotary.py:15-24). `t` is (..., d, m); `freqs`
    broadcasts over everything but the d axis."""
    rot_dim = freqs.shape[-2]
    t_rot, t_pass = t[..., :rot_dim, :], t[..., rot_dim:, :]
    # pair layout (2i, 2i+1) -> halves (-odd, even), as the reference's
    # rotate_half does (and as rope_P reproduces lane-wise in the kernel)
    even, odd = t_rot[..., 0::2, :], t_rot[..., 1::2, :]
    half_rot = torch.cat((-odd, even), dim=-2)
    t_rot = t_rot * freqs.cos() + half_rot * freqs.sin()
    return torch.cat((t_rot, t_pass), dim=-2)


def _first_tensor(features):
    return next(iter(features.values()))


def _attn_kernel_ok(q, J, dm):
    """Gate for the fused HIP attention kernel (csrc/attn2.hip): any J via
    online-softmax tiles; DM bounded by the per-lane register chunking."""
    from ..ops import fused as _fused
    max_j = int(os.environ.get('SE3_ATTN_MAX_J', 100000))
    return (q.is_cuda and J <= max_j and dm <= 448
            and q.dtype in (torch.float32, torch.bfloat16)
            and os.environ.get('SE3_EAGER_ATTN') != '1'
            and _fused.ext_available())


def _rope_in_kernel_ok(q, j_pre, qpe, kpe, use_null_kv, global_feats):
    """True if the rotary embedding can be folded into the attention kernel
    (q/k/v rotated in-registers) instead of materializing rotated copies.
    The frequency tables must not need grad (rotary_rel_dist with
    differentiable coordinates falls back to the eager path)."""
    rot = qpe.shape[-1]
    gj = 0
    if global_feats is not None:   # fiber dict {'0': (b, gj, d, 1)}
        gj = _first_tensor(global_feats).shape[1]
    j_fin = j_pre + (1 if use_null_kv else 0) + gj
    dm = q.shape[-2] * q.shape[-1]
    return (_attn_kernel_ok(q, j_fin, dm)
            and rot <= 64 and rot % 2 == 0 and rot <= dm
            and kpe.shape[2] == j_pre
            and not qpe.requires_grad and not kpe.requires_grad)


class AttentionSE3(nn.Module):
    def __init__(self, fiber, dim_head=64, heads=8, attend_self=False,
                 edge_dim=None, fourier_encode_dist=False,
                 rel_dist_num_fourier_features=4, use_null_kv=False, splits=4,
                 global_feats_dim=None, linear_proj_keys=False,
                 tie_key_values=False):
        super().__init__()
        hidden_dim = dim_head * heads
        hidden_fiber = Fiber([(degree, hidden_dim) for degree, _ in fiber])
        project_out = not (heads == 1 and len(fiber.dims) == 1 and dim_head == fiber.dims[0])

        self.scale = dim_head ** -0.5
        self.heads = heads
        self.linear_proj_keys = linear_proj_keys

        conv_kwargs = dict(edge_dim=edge_dim, pool=False, self_interaction=False,
                           fourier_encode_dist=fourier_encode_dist,
                           num_fourier_features=rel_dist_num_fourier_features,
                           splits=splits)

        self.to_q = LinearSE3(fiber, hidden_fiber)
        self.to_v = ConvSE3(fiber, hidden_fiber, **conv_kwargs)

        assert not (linear_proj_keys and tie_key_values), \
            'linear_proj_keys and tie_key_values are mutually exclusive'

        if linear_proj_keys:
            self.to_k = LinearSE3(fiber, hidden_fiber)
        elif not tie_key_values:
            self.to_k = ConvSE3(fiber, hidden_fiber, **conv_kwargs)
        else:
            self.to_k = None

        self.to_out = LinearSE3(hidden_fiber, fiber) if project_out else nn.Identity()

        self.use_null_kv = use_null_kv
        if use_null_kv:
            self.null_keys = nn.ParameterDict()
            self.null_values = nn.ParameterDict()
            for degree in fiber.degrees:
                m = to_order(degree)
                self.null_keys[str(degree)] = nn.Parameter(torch.zeros(heads, dim_head, m))
                self.null_values[str(degree)] = nn.Parameter(torch.zeros(heads, dim_head, m))

        self.attend_self = attend_self
        if attend_self:
            self.to_self_k = LinearSE3(fiber, hidden_fiber)
            self.to_self_v = LinearSE3(fiber, hidden_fiber)

        self.accept_global_feats = global_feats_dim is not None
        if self.accept_global_feats:
            global_input_fiber = Fiber.create(1, global_feats_dim)
            global_output_fiber = Fiber.create(1, hidden_fiber[0])
            self.to_global_k = LinearSE3(global_input_fiber, global_output_fiber)
            self.to_global_v = LinearSE3(global_input_fiber, global_output_fiber)

    def forward(self, features, edge_info, rel_dist, basis, global_feats=None,
                pos_emb=None, mask=None):
        h = self.heads
        neighbor_indices, neighbor_mask, edges = edge_info

        if neighbor_mask is not None:
            neighbor_mask = neighbor_mask.unsqueeze(1)  # b 1 i j

        queries = self.to_q(features)
        values = self.to_v(features, edge_info, rel_dist, basis)

        if self.linear_proj_keys:
            keys = self.to_k(features)
            keys = map_values(lambda v: batched_index_select(v, neighbor_indices, dim=1), keys)
        elif self.to_k is None:
            keys = values
        else:
            keys = self.to_k(features, edge_info, rel_dist, basis)

        if self.attend_self:
            self_keys, self_values = self.to_self_k(features), self.to_self_v(features)

        if global_feats is not None:
            global_keys, global_values = self.to_global_k(global_feats), self.to_global_v(global_feats)

        outputs = {}
        for degree in features.keys():
            q, k, v = queries[degree], keys[degree], values[degree]
            b, n = q.shape[0], q.shape[1]
            m = q.shape[-1]

            # b i (h d) m -> b h i d m
            q = q.view(b, n, h, -1, m).permute(0, 2, 1, 3, 4)
            # b i j (h d) m -> b h i j d m
            k = k.view(b, n, k.shape[2], h, -1, m).permute(0, 3, 1, 2, 4, 5)
            v = v.view(b, n, v.shape[2], h, -1, m).permute(0, 3, 1, 2, 4, 5)

            if self.attend_self:
                self_k, self_v = self_keys[degree], self_values[degree]
                self_k = self_k.view(b, n, h, -1, m).permute(0, 2, 1, 3, 4).unsqueeze(3)
                self_v = self_v.view(b, n, h, -1, m).permute(0, 2, 1, 3, 4).unsqueeze(3)
                k = torch.cat((self_k, k), dim=3)
                v = torch.cat((self_v, v), dim=3)

            rope_q = rope_k = None
            if pos_emb is not None and degree == '0':
                query_pos_emb, key_pos_emb = pos_emb
                if _rope_in_kernel_ok(q, k.shape[3], query_pos_emb, key_pos_emb,
                                      self.use_null_kv, global_feats):
                    # rotary folds into the attention kernel (q, k AND v are
                    # rotated there, reference :488-494) — no eager copies
                    rot_dim = query_pos_emb.shape[-1]
                    rope_q = query_pos_emb.reshape(-1, rot_dim).float().contiguous()
                    rope_k = key_pos_emb.reshape(-1, key_pos_emb.shape[2],
                                                 rot_dim).float().contiguous()
                else:
                    query_pos_emb = query_pos_emb[:, None, :, :, None]      # b 1 i d 1
                    key_pos_emb = key_pos_emb[:, None, :, :, :, None]       # b 1 i j d 1
                    q = apply_rotary_pos_emb(q, query_pos_emb)
                    k = apply_rotary_pos_emb(k, key_pos_emb)
                    v = apply_rotary_pos_emb(v, key_pos_emb)

            if self.use_null_kv:
                null_k, null_v = self.null_keys[degree], self.null_values[degree]
                null_k = null_k.view(1, h, 1, 1, -1, m).expand(b, h, n, 1, -1, m)
                null_v = null_v.view(1, h, 1, 1, -1, m).expand(b, h, n, 1, -1, m)
                k = torch.cat((null_k.to(k.dtype), k), dim=3)
                v = torch.cat((null_v.to(v.dtype), v), dim=3)

            if global_feats is not None and degree == '0':
                global_k, global_v = global_keys[degree], global_values[degree]
                gj = global_k.shape[1]
                global_k = global_k.view(b, gj, h, -1, m).permute(0, 2, 1, 3, 4)
                global_v = global_v.view(b, gj, h, -1, m).permute(0, 2, 1, 3, 4)
                global_k = global_k.unsqueeze(2).expand(b, h, k.shape[2], gj, global_k.shape[-2], m)
                global_v = global_v.unsqueeze(2).expand(b, h, v.shape[2], gj, global_v.shape[-2], m)
                k = torch.cat((global_k, k), dim=3)
                v = torch.cat((global_v, v), dim=3)

            bq, hq, nq, dq, mq = q.shape
            J = k.shape[3]
            from ..ops import fused as _fused
            if _attn_kernel_ok(q, J, dq * mq) or rope_q is not None:
                mask_u8 = None
                if neighbor_mask is not None:
                    pad = J - neighbor_mask.shape[-1]
                    mask_u8 = F.pad(neighbor_mask, (pad, 0), value=True) \
                        .squeeze(1).to(torch.uint8).contiguous()
                out = _fused.fused_attention(
                    q.reshape(bq * hq * nq, dq * mq).contiguous(),
                    k.to(q.dtype).reshape(bq * hq * nq, J, dq * mq).contiguous(),
                    v.to(q.dtype).reshape(bq * hq * nq, J, dq * mq).contiguous(),
                    mask_u8, nq, hq, self.scale, qf=rope_q, kf=rope_k)
                out = out.view(bq, hq, nq, dq, mq).to(q.dtype)
            else:
                sim = torch.einsum('bhidm,bhijdm->bhij', q, k) * self.scale

                if neighbor_mask is not None:
                    num_left_pad = sim.shape[-1] - neighbor_mask.shape[-1]
                    padded_mask = F.pad(neighbor_mask, (num_left_pad, 0), value=True)
                    sim = sim.masked_fill(~padded_mask, -torch.finfo(sim.dtype).max)

                attn = sim.softmax(dim=-1)
                out = torch.einsum('bhij,bhijdm->bhidm', attn, v)
            # b h n d m -> b n (h d) m
            outputs[degree] = out.permute(0, 2, 1, 3, 4).reshape(b, n, -1, m)

        return self.to_out(outputs)


class OneHeadedKVAttentionSE3(nn.Module):
    """AttentionSE3 with a single key/value head shared across query heads
    (reference :522-654; Shazeer arXiv 1911.02150)."""

    def __init__(self, fiber, dim_head=64, heads=8, attend_self=False,
                 edge_dim=None, fourier_encode_dist=False,
                 rel_dist_num_fourier_features=4, use_null_kv=False, splits=4,
                 global_feats_dim=None, linear_proj_keys=False,
                 tie_key_values=False):
        super().__init__()
        hidden_dim = dim_head * heads
        hidden_fiber = Fiber([(degree, hidden_dim) for degree, _ in fiber])
        kv_hidden_fiber = Fiber([(degree, dim_head) for degree, _ in fiber])
        project_out = not (heads == 1 and len(fiber.dims) == 1 and dim_head == fiber.dims[0])

        self.scale = dim_head ** -0.5
        self.heads = heads
        self.linear_proj_keys = linear_proj_keys

        conv_kwargs = dict(edge_dim=edge_dim, pool=False, self_interaction=False,
                           fourier_encode_dist=fourier_encode_dist,
                           num_fourier_features=rel_dist_num_fourier_features,
                           splits=splits)

        self.to_q = LinearSE3(fiber, hidden_fiber)
        self.to_v = ConvSE3(fiber, kv_hidden_fiber, **conv_kwargs)

        assert not (linear_proj_keys and tie_key_values)

        if linear_proj_keys:
            self.to_k = LinearSE3(fiber, kv_hidden_fiber)
        elif not tie_key_values:
            self.to_k = ConvSE3(fiber, kv_hidden_fiber, **conv_kwargs)
        else:
            self.to_k = None

        self.to_out = LinearSE3(hidden_fiber, fiber) if project_out else nn.Identity()

        self.use_null_kv = use_null_kv
        if use_null_kv:
            self.null_keys = nn.ParameterDict()
            self.null_values = nn.ParameterDict()
            for degree in fiber.degrees:
                m = to_order(degree)
                self.null_keys[str(degree)] = nn.Parameter(torch.zeros(dim_head, m))
                self.null_values[str(degree)] = nn.Parameter(torch.zeros(dim_head, m))

        self.attend_self = attend_self
        if attend_self:
            self.to_self_k = LinearSE3(fiber, kv_hidden_fiber)
            self.to_self_v = LinearSE3(fiber, kv_hidden_fiber)

        self.accept_global_feats = global_feats_dim is not None
        if self.accept_global_feats:
            global_input_fiber = Fiber.create(1, global_feats_dim)
            global_output_fiber = Fiber.create(1, kv_hidden_fiber[0])
            self.to_global_k = LinearSE3(global_input_fiber, global_output_fiber)
            self.to_global_v = LinearSE3(global_input_fiber, global_output_fiber)

    def forward(self, features, edge_info, rel_dist, basis, global_feats=None,
                pos_emb=None, mask=None):
        h = self.heads
        neighbor_indices, neighbor_mask, edges = edge_info

        if neighbor_mask is not None:
            neighbor_mask = neighbor_mask.unsqueeze(1)

        queries = self.to_q(features)
        values = self.to_v(features, edge_info, rel_dist, basis)

        if self.linear_proj_keys:
            keys = self.to_k(features)
            keys = map_values(lambda v: batched_index_select(v, neighbor_indices, dim=1), keys)
        elif self.to_k is None:
            keys = values
        else:
            keys = self.to_k(features, edge_info, rel_dist, basis)

        if self.attend_self:
            self_keys, self_values = self.to_self_k(features), self.to_self_v(features)

        if global_feats is not None:
            global_keys, global_values = self.to_global_k(global_feats), self.to_global_v(global_feats)

        outputs = {}
        for degree in features.keys():
            q, k, v = queries[degree], keys[degree], values[degree]
            b, n = q.shape[0], q.shape[1]
            m = q.shape[-1]

            q = q.view(b, n, h, -1, m).permute(0, 2, 1, 3, 4)  # b h i d m

            if self.attend_self:
                self_k, self_v = self_keys[degree], self_values[degree]
                k = torch.cat((self_k.unsqueeze(2), k), dim=2)
                v = torch.cat((self_v.unsqueeze(2), v), dim=2)

            rope_q = rope_k = None
            if pos_emb is not None and degree == '0':
                query_pos_emb, key_pos_emb = pos_emb
                if _rope_in_kernel_ok(q, k.shape[2], query_pos_emb, key_pos_emb,
                                      self.use_null_kv, global_feats):
                    rot_dim = query_pos_emb.shape[-1]
                    rope_q = query_pos_emb.reshape(-1, rot_dim).float().contiguous()
                    rope_k = key_pos_emb.reshape(-1, key_pos_emb.shape[2],
                                                 rot_dim).float().contiguous()
                else:
                    query_pos_emb = query_pos_emb[:, None, :, :, None]
                    key_pos_emb = key_pos_emb[..., None]
                    q = apply_rotary_pos_emb(q, query_pos_emb)
                    k = apply_rotary_pos_emb(k, key_pos_emb)
                    v = apply_rotary_pos_emb(v, key_pos_emb)

            if self.use_null_kv:
                null_k, null_v = self.null_keys[degree], self.null_values[degree]
                null_k = null_k.view(1, 1, 1, -1, m).expand(b, n, 1, -1, m)
                null_v = null_v.view(1, 1, 1, -1, m).expand(b, n, 1, -1, m)
                k = torch.cat((null_k.to(k.dtype), k), dim=2)
                v = torch.cat((null_v.to(v.dtype), v), dim=2)

            if global_feats is not None and degree == '0':
                global_k, global_v = global_keys[degree], global_values[degree]
                global_k = global_k.unsqueeze(1).expand(b, k.shape[1], global_k.shape[1], -1, m)
                global_v = global_v.unsqueeze(1).expand(b, v.shape[1], global_v.shape[1], -1, m)
                k = torch.cat((global_k, k), dim=2)
                v = torch.cat((global_v, v), dim=2)

            bq, hq, nq, dq, mq = q.shape
            J = k.shape[2]
            from ..ops import fused as _fused
            if _attn_kernel_ok(q, J, dq * mq) or rope_q is not None:
                mask_u8 = None
                if neighbor_mask is not None:
                    pad = J - neighbor_mask.shape[-1]
                    mask_u8 = F.pad(neighbor_mask, (pad, 0), value=True) \
                        .squeeze(1).to(torch.uint8).contiguous()
                out = _fused.fused_attention(
                    q.reshape(bq * hq * nq, dq * mq).contiguous(),
                    k.to(q.dtype).reshape(bq * nq, J, dq * mq).contiguous(),
                    v.to(q.dtype).reshape(bq * nq, J, dq * mq).contiguous(),
                    mask_u8, nq, hq, self.scale, True, qf=rope_q, kf=rope_k)
                out = out.view(bq, hq, nq, dq, mq).to(q.dtype)
            else:
                sim = torch.einsum('bhidm,bijdm->bhij', q, k) * self.scale

                if neighbor_mask is not None:
                    num_left_pad = sim.shape[-1] - neighbor_mask.shape[-1]
                    padded_mask = F.pad(neighbor_mask, (num_left_pad, 0), value=True)
                    sim = sim.masked_fill(~padded_mask, -torch.finfo(sim.dtype).max)

                attn = sim.softmax(dim=-1)
                out = torch.einsum('bhij,bijdm->bhidm', attn, v)
            outputs[degree] = out.permute(0, 2, 1, 3, 4).reshape(b, n, -1, m)

        return self.to_out(outputs)


class AttentionBlockSE3(nn.Module):
    def __init__(self, fiber, dim_head=24, heads=8, attend_self=False,
                 edge_dim=None, use_null_kv=False, fourier_encode_dist=False,
                 rel_dist_num_fourier_features=4, splits=4,
                 global_feats_dim=False, linear_proj_keys=False,
                 tie_key_values=False, attention_klass=AttentionSE3,
                 norm_gated_scale=False):
        super().__init__()
        self.attn = attention_klass(
            fiber, heads=heads, dim_head=dim_head, attend_self=attend_self,
            edge_dim=edge_dim, use_null_kv=use_null_kv,
            rel_dist_num_fourier_features=rel_dist_num_fourier_features,
            fourier_encode_dist=fourier_encode_dist, splits=splits,
            global_feats_dim=global_feats_dim, linear_proj_keys=linear_proj_keys,
            tie_key_values=tie_key_values)
        self.prenorm = NormSE3(fiber, gated_scale=norm_gated_scale)
        self.residual = ResidualSE3()

    def forward(self, features, edge_info, rel_dist, basis, global_feats=None,
                pos_emb=None, mask=None):
        res = features
        outputs = self.prenorm(features)
        outputs = self.attn(outputs, edge_info, rel_dist, basis, global_feats, pos_emb, mask)
        return self.residual(outputs, res)
