"""Reversible (RevNet-style) trunk: O(1) activation memory, inputs recomputed
by inversion in backward (Gomez et al., arXiv 1707.04585).

Module tree parity with reference reversible.py (ReversibleBlock holds
f = Deterministic(attention block) and g = Deterministic(ff block) so the
state-dict prefixes are net.blocks.N.f.net.* / net.blocks.N.g.net.*).
Per-device RNG capture/restore keeps recomputation deterministic.
"""
from __future__ import annotations

import torch
import torch.nn as nn
from torch.autograd.function import Function
from torch.utils.checkpoint import get_device_states, set_device_states

from ..utils import map_values


def _dict_chunk2(x, dim):
    out1, out2 = {}, {}
    for k, v in x.items():
        c1, c2 = v.chunk(2, dim=dim)
        out1[k], out2[k] = c1, c2
    return out1, out2


def _dict_sum(x, y):
    return {k: x[k] + y[k] for k in x}


def _dict_sub(x, y):
    return {k: x[k] - y[k] for k in x}


def _dict_cat(x, y, dim):
    return {k: torch.cat((x[k], y[k]), dim=dim) for k in x}


class Deterministic(nn.Module):
    """Wrap a module, recording RNG state at forward so a later re-forward
    (set_rng=True) reproduces it exactly."""

    def __init__(self, net):
        super().__init__()
        self.net = net
        self.cpu_state = None
        self.cuda_in_fwd = None
        self.gpu_devices = None
        self.gpu_states = None

    def record_rng(self, *args):
        self.cpu_state = torch.get_rng_state()
        if torch.cuda._initialized:
            self.cuda_in_fwd = True
            self.gpu_devices, self.gpu_states = get_device_states(*args)

    def forward(self, *args, record_rng=False, set_rng=False, **kwargs):
        if record_rng:
            self.record_rng(*args)
        if not set_rng:
            return self.net(*args, **kwargs)
        rng_devices = self.gpu_devices if self.cuda_in_fwd else []
        with torch.random.fork_rng(devices=rng_devices, enabled=True):
            torch.set_rng_state(self.cpu_state)
            if self.cuda_in_fwd:
                set_device_states(self.gpu_devices, self.gpu_states)
            return self.net(*args, **kwargs)


class ReversibleBlock(nn.Module):
    """y1 = x1 + f(x2); y2 = x2 + g(y1) over fiber dicts split along channels."""

    def __init__(self, f, g):
        super().__init__()
        self.f = Deterministic(f)
        self.g = Deterministic(g)

    def forward(self, x, **kwargs):
        training = self.training
        x1, x2 = _dict_chunk2(x, dim=-1)
        with torch.no_grad():
            y1 = _dict_sum(x1, self.f(x2, record_rng=training, **kwargs))
            y2 = _dict_sum(x2, self.g(y1, record_rng=training))
        return _dict_cat(y1, y2, dim=-1)

    def backward_pass(self, y, dy, **kwargs):
        y1, y2 = _dict_chunk2(y, dim=-1)
        dy1, dy2 = _dict_chunk2(dy, dim=-1)

        with torch.enable_grad():
            y1 = map_values(lambda t: t.detach().requires_grad_(True), y1)
            gy1 = self.g(y1, set_rng=True)
            torch.autograd.backward(list(gy1.values()), list(dy2.values()))

        with torch.no_grad():
            x2 = _dict_sub(y2, gy1)
            dx1 = _dict_sum(dy1, map_values(lambda t: t.grad, y1))

        with torch.enable_grad():
            x2 = map_values(lambda t: t.detach().requires_grad_(True), x2)
            fx2 = self.f(x2, set_rng=True, **kwargs)
            torch.autograd.backward(list(fx2.values()), list(dx1.values()))

        with torch.no_grad():
            x1 = _dict_sub({k: v.detach() for k, v in y1.items()}, fx2)
            dx2 = _dict_sum(dy2, map_values(lambda t: t.grad, x2))
            x = _dict_cat(x1, map_values(lambda t: t.detach(), x2), dim=-1)
            dx = _dict_cat(dx1, dx2, dim=-1)
        return x, dx


class _ReversibleFunction(Function):
    @staticmethod
    def forward(ctx, x, blocks, kwargs):
        input_keys = kwargs.pop('input_keys')
        split_dims = kwargs.pop('split_dims')
        x = dict(zip(input_keys, x.split(split_dims, dim=-1)))

        ctx.kwargs = kwargs
        ctx.split_dims = split_dims
        ctx.input_keys = input_keys

        for block in blocks:
            x = block(x, **kwargs)

        ctx.y = map_values(lambda t: t.detach(), x)
        ctx.blocks = blocks
        return torch.cat(list(x.values()), dim=-1)

    @staticmethod
    def backward(ctx, dy):
        y = ctx.y
        dy = dict(zip(ctx.input_keys, dy.split(ctx.split_dims, dim=-1)))
        for block in ctx.blocks[::-1]:
            y, dy = block.backward_pass(y, dy, **ctx.kwargs)
        dy = torch.cat(list(dy.values()), dim=-1)
        return dy, None, None


class SequentialSequence(nn.Module):
    def __init__(self, blocks):
        super().__init__()
        self.blocks = blocks

    def forward(self, x, **kwargs):
        for attn, ff in self.blocks:
            x = attn(x, **kwargs)
            x = ff(x)
        return x


class ReversibleSequence(nn.Module):
    """Duplicates the fiber channels, runs reversible blocks, averages halves.

    Layout note: duplication and the x1/x2 split are on dim=-1 (the m dim,
    as reference reversible.py:208): the split lands exactly on the
    duplication boundary, so each half is a full, equivariant fiber tensor.
    """

    def __init__(self, blocks):
        super().__init__()
        self.blocks = nn.ModuleList([ReversibleBlock(f, g) for f, g in blocks])

    def forward(self, x, **kwargs):
        x = map_values(lambda t: torch.cat((t, t), dim=-1), x)
        input_keys = list(x.keys())
        split_dims = tuple(t.shape[-1] for t in x.values())
        block_kwargs = {'input_keys': input_keys, 'split_dims': split_dims, **kwargs}
        flat = torch.cat(list(x.values()), dim=-1)
        flat = _ReversibleFunction.apply(flat, self.blocks, block_kwargs)
        x = dict(zip(input_keys, flat.split(split_dims, dim=-1)))
        return map_values(lambda t: torch.stack(t.chunk(2, dim=-1)).mean(dim=0), x)
