"""Dual torch/numpy backend plumbing for the geometry toolbox.

Capability parity with the reference's decorator discipline
(reference utils.py:54-104): every public geometry wrapper accepts
``backend='auto'|'torch'|'numpy'`` and dispatches to a torch or numpy
twin.  This keeps the whole geometry layer unit-testable on CPU without
a GPU in the loop.
"""
import contextlib
from functools import wraps

import torch


def exists(val):
    return val is not None


def expand_dims_to(t, length=3):
    """Left-pad the shape of a tensor/array with singleton dims."""
    if length <= 0:
        return t
    return t.reshape(*((1,) * length), *t.shape)


def set_backend_kwarg(fn):
    """Resolve backend='auto' from the type of the first argument."""
    @wraps(fn)
    def inner(*args, backend='auto', **kwargs):
        if backend == 'auto':
            backend = 'torch' if isinstance(args[0], torch.Tensor) else 'numpy'
        kwargs.update(backend=backend)
        return fn(*args, **kwargs)
    return inner


def expand_arg_dims(dim_len=3):
    """Expand the two positional args to `dim_len` dims (adds batch dims)."""
    def outer(fn):
        @wraps(fn)
        def inner(x, y, **kwargs):
            assert len(x.shape) == len(y.shape), "shapes of A and B must match"
            remaining = dim_len - len(x.shape)
            x = expand_dims_to(x, length=remaining)
            y = expand_dims_to(y, length=remaining)
            return fn(x, y, **kwargs)
        return inner
    return outer


def invoke_torch_or_numpy(torch_fn, numpy_fn):
    """The wrapped fn returns the positional args (and optionally a kwargs
    dict as last element); the selected backend twin is then invoked."""
    def outer(fn):
        @wraps(fn)
        def inner(*args, **kwargs):
            backend = kwargs.pop('backend')
            passed_args = list(fn(*args, **kwargs))
            if isinstance(passed_args[-1], dict):
                passed_kwargs = passed_args.pop()
            else:
                passed_kwargs = {}
            backend_fn = torch_fn if backend == 'torch' else numpy_fn
            return backend_fn(*passed_args, **passed_kwargs)
        return inner
    return outer


@contextlib.contextmanager
def torch_default_dtype(dtype):
    prev = torch.get_default_dtype()
    torch.set_default_dtype(dtype)
    try:
        yield
    finally:
        torch.set_default_dtype(prev)
