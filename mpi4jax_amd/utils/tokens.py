"""Token API parity with the reference.

The reference removed user-visible tokens in 0.8.0 and raises when one is
passed (``/root/reference/mpi4jax/_src/utils.py:30-42``); ordering is instead
enforced internally by a JAX ordered effect (token threading,
``utils.py:45-53``).  Here the same guarantee — program order == network
order per communicator — is enforced in hardware: every collective is
enqueued on the current compute stream (program order == stream order),
and when consecutive collectives land on *different* HIP streams the
backend bridges them with a stream event
(``_backend/rccl._order_fence``), so not even multi-stream code can
reorder a communicator's sequence against another rank's.  We keep the
``token=NOTSET`` kwarg and raise the same way for drop-in compatibility.
"""


class _NotSetType:
    _instance = None

    def __new__(cls):
        if cls._instance is None:
            cls._instance = super().__new__(cls)
        return cls._instance

    def __repr__(self):
        return "NOTSET"

    def __bool__(self):
        return False


NOTSET = _NotSetType()


class Token:
    """Opaque ordering token (compatibility object).

    Exists only so code written against very old reference versions that
    manipulated tokens has something to hold; carrying it has no effect —
    ordering is provided by HIP stream order.
    """

    __slots__ = ()

    def __repr__(self):
        return "Token()"


def raise_if_token_is_set(token):
    if token is NOTSET:
        return
    # mirror the reference's messaging (utils.py:30-42)
    raise RuntimeError(
        "mpi4jax_amd does not accept explicit tokens. Communication is "
        "ordered automatically by HIP stream order; remove the `token` "
        "argument. (The reference mpi4jax removed explicit tokens in "
        "version 0.8.0 for the same reason.)"
    )
