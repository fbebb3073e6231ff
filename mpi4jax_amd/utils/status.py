"""MPI-like Status object and wildcard constants.

The reference passes ``mpi4py.MPI.Status`` pointers into its custom calls
(``/root/reference/mpi4jax/_src/collective_ops/recv.py:100-112``).  RCCL has
no message envelope, so ``recv``/``sendrecv`` here synthesize the fields
(shapes are static anyway, exactly as noted in SURVEY.md §2.3 #11).
"""

ANY_SOURCE = -1
ANY_TAG = -1


class Status:
    """Introspection object filled in by :func:`recv` / :func:`sendrecv`.

    Mirrors the mpi4py surface actually used by the reference tests:
    ``Get_source()`` / ``.source`` and ``Get_tag()`` / ``.tag``.
    """

    def __init__(self):
        self.source = ANY_SOURCE
        self.tag = ANY_TAG
        self.count = 0

    def Get_source(self):
        return self.source

    def Get_tag(self):
        return self.tag

    def Get_count(self):
        return self.count

    def __repr__(self):
        return f"Status(source={self.source}, tag={self.tag}, count={self.count})"
