"""Runtime type validation for static (non-tensor) op arguments.

Analog of the reference's ``@enforce_types`` decorator
(``/root/reference/mpi4jax/_src/validation.py:8-94``), including the special
error message when a tensor leaks into a static argument (the reference's
version of this catches JAX tracers, :77-88).
"""

import functools
import inspect

import torch


def _typename(t):
    if isinstance(t, tuple):
        return " or ".join(_typename(x) for x in t)
    return getattr(t, "__name__", str(t))


def enforce_types(**type_specs):
    """Decorator enforcing that named keyword/positional args match a spec.

    Spec values are a type or tuple of types.  ``int`` specs additionally
    accept any Python/numpy integer.
    """

    def wrap(fn):
        sig = inspect.signature(fn)

        @functools.wraps(fn)
        def wrapped(*args, **kwargs):
            bound = sig.bind(*args, **kwargs)
            for name, spec in type_specs.items():
                if name not in bound.arguments:
                    continue
                val = bound.arguments[name]
                if isinstance(val, torch.Tensor):
                    raise TypeError(
                        f"{fn.__name__}: argument '{name}' must be static "
                        f"(got a torch.Tensor). Tensor-valued static "
                        f"arguments are not supported — pass a Python "
                        f"{_typename(spec)} instead."
                    )
                if spec is int or (isinstance(spec, tuple) and int in spec):
                    # accept numpy integer scalars as well
                    if hasattr(val, "__index__"):
                        continue
                if not isinstance(val, spec):
                    raise TypeError(
                        f"{fn.__name__}: expected '{name}' to be of type "
                        f"{_typename(spec)}, got {type(val).__name__}"
                    )
            return fn(*args, **kwargs)

        return wrapped

    return wrap
