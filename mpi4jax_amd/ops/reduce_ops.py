"""Reduction operator enum.

The reference takes ``mpi4py.MPI.Op`` objects (``MPI.SUM`` etc.) and smuggles
their handles into XLA custom-call attributes
(``/root/reference/mpi4jax/_src/utils.py:60-90``).  This framework is
standalone: it defines its own :class:`Op` enum; if mpi4py happens to be
installed, its ops are transparently mapped for drop-in compatibility.

SUM/PROD/MIN/MAX (and AVG, an RCCL extra) run natively on both backends.
The bitwise ops BAND/BOR/BXOR (integer/bool dtypes) run on gloo natively
and on the GPU through the HIP combine kernel over p2p chains (RCCL has
no bitwise reductions): scan is the ring it already uses, and
allreduce/reduce compose scan with a broadcast of the final prefix.
"""

import enum

import torch
import torch.distributed as dist


class Op(enum.Enum):
    SUM = "sum"
    PROD = "prod"
    MIN = "min"
    MAX = "max"
    AVG = "avg"  # RCCL-native extra (not in MPI)
    BAND = "band"
    BOR = "bor"
    BXOR = "bxor"

    def __repr__(self):
        return f"Op.{self.name}"


SUM = Op.SUM
PROD = Op.PROD
MIN = Op.MIN
MAX = Op.MAX
AVG = Op.AVG
BAND = Op.BAND
BOR = Op.BOR
BXOR = Op.BXOR

# rccl.h ncclRedOp_t values
RCCL_OP_ENUM = {
    Op.SUM: 0,
    Op.PROD: 1,
    Op.MAX: 2,
    Op.MIN: 3,
    Op.AVG: 4,
}

# kernel-only codes for the HIP combine kernel (csrc/kernels.h OpCode):
# RCCL has no bitwise reductions, so on the GPU these ops ride the
# p2p-chain compositions (scan ring; allreduce/reduce = scan + bcast)
# and never reach an RCCL collective.
BITWISE_OP_ENUM = {
    Op.BAND: 5,
    Op.BOR: 6,
    Op.BXOR: 7,
}

GLOO_OP_MAP = {
    Op.SUM: dist.ReduceOp.SUM,
    Op.PROD: dist.ReduceOp.PRODUCT,
    Op.MIN: dist.ReduceOp.MIN,
    Op.MAX: dist.ReduceOp.MAX,
    Op.BAND: dist.ReduceOp.BAND,
    Op.BOR: dist.ReduceOp.BOR,
    Op.BXOR: dist.ReduceOp.BXOR,
}


def resolve_op(op, op_name="reduction"):
    """Accept an Op, its string name (as jit_ops does), or an mpi4py Op
    if mpi4py is importable."""
    if isinstance(op, Op):
        return op
    if isinstance(op, str):
        try:
            return Op(op.lower())
        except ValueError:
            pass  # fall through to the uniform TypeError below
    # optional mpi4py compatibility
    try:
        from mpi4py import MPI  # noqa

        mapping = {
            MPI.SUM: Op.SUM,
            MPI.PROD: Op.PROD,
            MPI.MIN: Op.MIN,
            MPI.MAX: Op.MAX,
            MPI.BAND: Op.BAND,
            MPI.BOR: Op.BOR,
            MPI.BXOR: Op.BXOR,
        }
        if op in mapping:
            return mapping[op]
    except ImportError:
        pass
    raise TypeError(
        f"{op_name}: expected an mpi4jax_amd.Op (e.g. mpi4jax_amd.SUM), "
        f"got {op!r}"
    )


def combine(op: Op, a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Local elementwise combine used by the CPU scan chain."""
    if op is Op.SUM or op is Op.AVG:
        return a + b
    if op is Op.PROD:
        return a * b
    if op is Op.MIN:
        return torch.minimum(a, b)
    if op is Op.MAX:
        return torch.maximum(a, b)
    if op is Op.BAND:
        return a & b
    if op is Op.BOR:
        return a | b
    if op is Op.BXOR:
        return a ^ b
    raise ValueError(f"cannot combine with {op}")
