"""Rotary positional embedding tables for degree-0 attention channels.

On the MI355X path the rotation itself happens INSIDE the attention kernel
(csrc/attn2.hip rotates q/k/v in-registers from the raw frequency tables),
so this module is only the host-side table builder. The eager fallback
rotation lives next to its consumer in models/attention.py
(`apply_rotary_pos_emb`, re-exported here for reference API parity with
rotary.py:1-24).
"""
from __future__ import annotations

import torch
from torch import nn


class SinusoidalEmbeddings(nn.Module):
    """Frequency table: t -> (…, dim) with pairwise-repeated inverse
    frequencies (reference rotary.py:5-13)."""

    def __init__(self, dim):
        super().__init__()
        inv_freq = 1. / (10000 ** (torch.arange(0, dim, 2).float() / dim))
        self.register_buffer('inv_freq', inv_freq)

    def forward(self, t):
        freqs = t[..., None].float() * self.inv_freq[None, :]
        return freqs.repeat_interleave(2, dim=-1)


def apply_rotary_pos_emb(t, freqs):
    from .attention import apply_rotary_pos_emb as _impl
    return _impl(t, freqs)
