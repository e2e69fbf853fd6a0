"""Fiber: the degree -> channel-count structure of equivariant features.

API parity with reference se3_transformer_pytorch.py:18-59.
"""
from __future__ import annotations

from collections import namedtuple
from itertools import product

from torch import nn

FiberEl = namedtuple('FiberEl', ['degrees', 'dim'])


class Fiber(nn.Module):
    def __init__(self, structure):
        super().__init__()
        if isinstance(structure, dict):
            structure = [FiberEl(degree, dim) for degree, dim in structure.items()]
        self.structure = [FiberEl(*el) for el in structure]

    @property
    def dims(self):
        return list(dict.fromkeys(el.dim for el in self.structure))

    @property
    def degrees(self):
        # a fresh iterator each call, matching the reference's map() contract
        return iter([el.degrees for el in self.structure])

    @staticmethod
    def create(num_degrees, dim):
        dims = dim if isinstance(dim, tuple) else (dim,) * num_degrees
        return Fiber([FiberEl(d, dims[d]) for d in range(num_degrees)])

    def __getitem__(self, degree):
        return dict(self.structure)[degree]

    def __iter__(self):
        return iter(self.structure)

    def __mul__(self, other):
        return product(self.structure, other.structure)

    def __and__(self, other):
        out = []
        other_degrees = set(el.degrees for el in other.structure)
        for degree, dim in self:
            if degree in other_degrees:
                out.append((degree, dim, other[degree]))
        return out
