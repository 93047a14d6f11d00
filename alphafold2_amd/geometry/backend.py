"""Backend dispatch for the geometry toolbox.

The reference expresses its torch/numpy dual backend as three stacked
decorators (utils.py:54-104: backend-kwarg resolution, positional-arg
dim expansion, twin invocation).  Here the whole policy lives in ONE
factory, `dual_backend`: a public wrapper is built from its torch and
numpy implementations plus an optional argument-preparation hook, so
each geometry entry point is a single declaration instead of a
decorator tower.  The capability is the same — every public wrapper
accepts ``backend='auto'|'torch'|'numpy'`` and stays unit-testable on
CPU — but the mechanism is this package's own.
"""
import contextlib
from functools import wraps

import torch


def exists(val):
    return val is not None


def as_batched(t, ndim):
    """View a tensor/array with leading singleton axes so it has `ndim`
    dims (no-op when it already does)."""
    shape = t.shape
    if len(shape) >= ndim:
        return t
    return t.reshape((1,) * (ndim - len(shape)) + tuple(shape))


# transitional alias for the reference-style call convention
# (length = how many leading singleton dims to ADD)
def expand_dims_to(t, length=3):
    if length <= 0:
        return t
    return as_batched(t, len(t.shape) + length)


def resolve_backend(sample, requested='auto'):
    """'auto' means: torch when the probe argument is a torch tensor,
    numpy otherwise."""
    if requested != 'auto':
        return requested
    return 'torch' if isinstance(sample, torch.Tensor) else 'numpy'


def dual_backend(torch_impl, numpy_impl, pair_ndim=None, prepare=None):
    """Build a public geometry wrapper from two backend twins.

    * pair_ndim: when set, the first two positional args are promoted to
      this rank (adds batch dims) and must agree in rank.
    * prepare(*args, **kwargs) -> (args, kwargs): optional hook that
      normalizes/augments the arguments before the twin is invoked.

    The returned wrapper takes ``backend='auto'|'torch'|'numpy'``.
    """
    def dispatch(*args, backend='auto', **kwargs):
        impl = torch_impl if resolve_backend(args[0], backend) == 'torch' \
            else numpy_impl
        if pair_ndim is not None:
            a, b = args[0], args[1]
            assert len(a.shape) == len(b.shape), \
                'the two inputs must have the same rank'
            args = (as_batched(a, pair_ndim), as_batched(b, pair_ndim)) \
                + args[2:]
        if prepare is not None:
            args, kwargs = prepare(*args, **kwargs)
        return impl(*args, **kwargs)
    return dispatch


def named_wrapper(fn, name, doc):
    """Give a dual_backend dispatcher a public name + docstring."""
    fn.__name__ = name
    fn.__qualname__ = name
    fn.__doc__ = doc
    return fn


@contextlib.contextmanager
def torch_default_dtype(dtype):
    """Standard default-dtype scope (the canonical try/finally idiom)."""
    prev = torch.get_default_dtype()
    torch.set_default_dtype(dtype)
    try:
        yield
    finally:
        torch.set_default_dtype(prev)
