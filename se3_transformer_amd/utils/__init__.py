"""Generic tensor utilities for the MI355X-native SE(3)-Transformer framework.

Functional parity targets (reference: /root/reference/se3_transformer_pytorch/utils.py,
lines cited per function) but all implementations are written fresh and kept
functional (no in-place mutation of caller tensors, unlike the reference's
masked_mean at utils.py:75).
"""
from __future__ import annotations

import contextlib
import time
from functools import lru_cache, wraps

import torch


def exists(val):
    return val is not None


def default(val, d):
    return val if val is not None else d


def uniq(arr):
    return list(dict.fromkeys(arr))


def to_order(degree: int) -> int:
    """Dimension of the degree-l irrep: 2l+1 (reference utils.py:24)."""
    return 2 * degree + 1


def degree_total(max_degree: int) -> int:
    """Total SH components for degrees 0..max_degree = (max_degree+1)^2."""
    return (max_degree + 1) ** 2


def map_values(fn, d):
    return {k: fn(v) for k, v in d.items()}


def safe_cat(acc, el, dim):
    if acc is None:
        return el
    return torch.cat((acc, el), dim=dim)


def cast_tuple(val, depth):
    return val if isinstance(val, tuple) else (val,) * depth


def rand_uniform(size, min_val, max_val):
    return torch.empty(size).uniform_(min_val, max_val)


def broadcat(tensors, dim=-1):
    """Concatenate along `dim`, broadcasting all other (size-1) dims.

    Same contract as reference utils.py:38-54.
    """
    ndims = {t.dim() for t in tensors}
    assert len(ndims) == 1, 'tensors must all have the same number of dimensions'
    nd = ndims.pop()
    if dim < 0:
        dim += nd
    # target shape per non-cat dim is the max over tensors
    target = []
    for i in range(nd):
        if i == dim:
            target.append(-1)
            continue
        sizes = {t.shape[i] for t in tensors}
        sizes.discard(1)
        assert len(sizes) <= 1, 'invalid dimensions for broadcastable concatenation'
        target.append(sizes.pop() if sizes else 1)
    expanded = []
    for t in tensors:
        shape = [target[i] if i != dim else t.shape[i] for i in range(nd)]
        expanded.append(t.expand(*shape))
    return torch.cat(expanded, dim=dim)


def batched_index_select(values, indices, dim=1):
    """Gather along `dim` with per-batch indices that may add extra dims.

    values:  (*batch, N, *value_dims)  where len(batch) == dim
    indices: (*batch, *extra)          integer indices into N
    returns: (*batch, *extra, *value_dims)

    Same contract as reference utils.py:56-70; fresh gather-based impl.
    """
    batch_shape = values.shape[:dim]
    value_dims = values.shape[dim + 1:]
    extra = indices.shape[len(batch_shape):]

    flat_idx = indices.reshape(*batch_shape, -1)
    idx = flat_idx.reshape(*flat_idx.shape, *((1,) * len(value_dims)))
    idx = idx.expand(*batch_shape, flat_idx.shape[-1], *value_dims)
    out = values.gather(dim, idx)
    return out.reshape(*batch_shape, *extra, *value_dims)


def masked_mean(tensor, mask, dim=-1):
    """Mean over `dim` counting only entries where mask is True.

    mask has the leading dims of tensor; trailing dims broadcast.
    Functional (does NOT mutate `tensor`, unlike reference utils.py:75).
    """
    if mask is None:
        return tensor.mean(dim=dim)
    diff = tensor.dim() - mask.dim()
    mask = mask.reshape(*mask.shape, *((1,) * diff))
    t = tensor.masked_fill(~mask, 0.)
    total = mask.sum(dim=dim)
    mean = t.sum(dim=dim) / total.clamp(min=1.)
    return mean.masked_fill(total == 0, 0.)


def fourier_encode(x, num_encodings=4, include_self=True, flatten=True):
    """Multi-scale sin/cos encoding (reference utils.py:96-104 contract)."""
    x = x.unsqueeze(-1)
    orig_x = x
    scales = 2 ** torch.arange(num_encodings, device=x.device, dtype=x.dtype)
    x = x / scales
    x = torch.cat([x.sin(), x.cos()], dim=-1)
    if include_self:
        x = torch.cat((x, orig_x), dim=-1)
    if flatten:
        x = x.reshape(*x.shape[:3], -1)
    return x


@contextlib.contextmanager
def torch_default_dtype(dtype):
    prev = torch.get_default_dtype()
    torch.set_default_dtype(dtype)
    try:
        yield
    finally:
        torch.set_default_dtype(prev)


def cast_torch_tensor(fn):
    @wraps(fn)
    def inner(t):
        if not torch.is_tensor(t):
            t = torch.tensor(t, dtype=torch.get_default_dtype())
        return fn(t)
    return inner


def benchmark(fn):
    """Wall-clock wrapper returning (seconds, result)."""
    def inner(*args, **kwargs):
        start = time.time()
        res = fn(*args, **kwargs)
        return time.time() - start, res
    return inner


def fast_split(arr, splits, dim=0):
    """Near-equal chunking generator (API compat; reference utils.py:85-94).

    Kept for API parity only — the MI355X compute path streams edges inside
    fused kernels instead of chunking at the Python level.
    """
    n = arr.shape[dim]
    splits = min(n, max(splits, 1))
    base, rem = divmod(n, splits)
    start = 0
    for i in range(splits):
        size = base + (1 if i < rem else 0)
        yield torch.narrow(arr, dim, start, size)
        start += size


def cache(cache_dict, key_fn):
    """In-RAM memoization with a custom key function (reference
    utils.py:135-147)."""
    def cache_inner(fn):
        @wraps(fn)
        def inner(*args, **kwargs):
            key = key_fn(*args, **kwargs)
            if key in cache_dict:
                return cache_dict[key]
            res = fn(*args, **kwargs)
            cache_dict[key] = res
            return res
        return inner
    return cache_inner


def cache_dir(dirname, maxsize=128):
    """On-disk cache decorator (reference utils.py:151-206). Keeps the
    reference contract (results keyed by repr of the call, persisted under
    `dirname`, lru_cache front, CLEAR_CACHE env disables) but uses atomic
    tmp+rename writes instead of a FileLock."""
    import hashlib
    import os
    import tempfile

    import torch as _torch

    def decorator(func):
        if os.environ.get('CLEAR_CACHE') is not None:
            return func
        root = os.path.expanduser(dirname)

        @lru_cache(maxsize=maxsize)
        @wraps(func)
        def wrapper(*args, **kwargs):
            key = repr((args, tuple(sorted(kwargs.items()))))
            h = hashlib.sha1(key.encode()).hexdigest()[:24]
            path = os.path.join(root, f'{func.__name__}.{h}.pt')
            if os.path.exists(path):
                try:
                    # cached values are tensors/tuples of tensors; refuse
                    # arbitrary pickles from the user-writable cache dir
                    return _torch.load(path, weights_only=True)
                except Exception:
                    pass
            result = func(*args, **kwargs)
            try:
                os.makedirs(root, exist_ok=True)
                fd, tmp = tempfile.mkstemp(dir=root)
                os.close(fd)
                _torch.save(result, tmp)
                os.replace(tmp, path)
            except OSError:
                pass
            return result
        return wrapper
    return decorator
