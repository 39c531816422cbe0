"""Small helpers (reference utils.py had cost-model tables + misc; the
cost model lives in solver.py here)."""
import os


def create_path(path):
    os.makedirs(path, exist_ok=True)
    return path


def check_unique(seq):
    return len(set(seq)) == len(seq)


def force_insert_item(d, key, val):
    d.setdefault(key, []).append(val)


def is_dense(t):
    """True when the tensor's storage is a bijective dense permutation of
    its elements (contiguous in SOME order, e.g. channels_last)."""
    if t.is_contiguous():
        return True
    n = t.numel()
    span = 1 + sum((s - 1) * st for s, st in zip(t.shape, t.stride())
                   if s > 1)
    return span == n


def dense_flat_alias(t):
    """1-D alias of a dense tensor's raw storage order (no copy).

    For channels_last parameters this gives the NHWC byte order — the
    order the bucket views, fused SGD tables and collectives all share,
    so no transposes appear anywhere on the hot path.
    """
    if t.is_contiguous():
        return t.view(-1)
    if not is_dense(t):
        raise ValueError('tensor is not dense; cannot alias storage')
    return t.as_strided((t.numel(),), (1,))


def grad_view_like(flat_slice, p):
    """View a flat-buffer slice with p's logical shape AND strides, so
    autograd accumulates in p's native memory format (channels_last
    conv weights get NHWC gradient accumulation with zero transposes)."""
    if p.is_contiguous():
        return flat_slice.view_as(p)
    if not is_dense(p):
        raise ValueError('parameter is not dense')
    return flat_slice.as_strided(p.shape, p.stride())
