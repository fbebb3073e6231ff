"""Tensor-parallel layers: sharded forward+backward must match the dense
reference computation (the Megatron f/g pattern on our collectives)."""

import torch

import mpi4jax_amd as m
from tests._mp import run_multiproc


def _tp_mlp(rank, ws):
    from mpi4jax_amd.parallel import (ColumnParallelLinear,
                                      RowParallelLinear)

    torch.manual_seed(0)  # same dense weights everywhere
    din, dhid, dout, B = 8, 12, 6, 4
    dense1 = torch.nn.Linear(din, dhid).double()
    dense2 = torch.nn.Linear(dhid, dout).double()
    x = torch.randn(B, din, dtype=torch.float64, requires_grad=True)

    # reference dense forward/backward
    ref = dense2(torch.tanh(dense1(x)))
    ref.sum().backward()
    ref_xgrad, x.grad = x.grad.clone(), None

    # TP pair: column-parallel (sharded hidden) -> row-parallel
    col = ColumnParallelLinear(din, dhid, dtype=torch.float64)
    row = RowParallelLinear(dhid, dout, dtype=torch.float64)
    sh = slice(rank * dhid // ws, (rank + 1) * dhid // ws)
    with torch.no_grad():
        col.linear.weight.copy_(dense1.weight[sh])
        col.linear.bias.copy_(dense1.bias[sh])
        row.linear.weight.copy_(dense2.weight[:, sh])
        row.bias.copy_(dense2.bias)

    y = row(torch.tanh(col(x)))
    assert torch.allclose(y, ref, atol=1e-12), (y - ref).abs().max()
    y.sum().backward()
    assert torch.allclose(x.grad, ref_xgrad, atol=1e-12)
    # weight grads match the dense layer's corresponding shard
    assert torch.allclose(col.linear.weight.grad, dense1.weight.grad[sh],
                          atol=1e-12)
    assert torch.allclose(row.linear.weight.grad,
                          dense2.weight.grad[:, sh], atol=1e-12)


def test_tp_mlp_2ranks():
    run_multiproc(_tp_mlp, 2)


def test_tp_single_rank():
    from mpi4jax_amd.parallel import (ColumnParallelLinear,
                                      RowParallelLinear, copy_to_parallel)

    x = torch.randn(3, 4, requires_grad=True)
    y = copy_to_parallel(x)
    y.sum().backward()
    assert torch.equal(x.grad, torch.ones_like(x))
    col = ColumnParallelLinear(4, 6, gather_output=True)
    row = RowParallelLinear(6, 2)
    out = row(col(x.detach()))
    assert out.shape == (3, 2)


def _ulysses_roundtrip(rank, ws):
    from mpi4jax_amd.parallel import seq_to_head_shard, head_to_seq_shard

    S, H, D = 4 * ws, 2 * ws, 3
    torch.manual_seed(5)
    full = torch.randn(S, H, D)  # the logical global tensor
    mine_seq = full[rank * S // ws:(rank + 1) * S // ws]  # (S/P, H, D)

    heads = seq_to_head_shard(mine_seq)  # (S, H/P, D)
    expect_heads = full[:, rank * H // ws:(rank + 1) * H // ws]
    assert torch.equal(heads, expect_heads), (rank, "to-heads")

    back = head_to_seq_shard(heads)
    assert torch.equal(back, mine_seq), (rank, "roundtrip")


def test_ulysses_shard_swap_2ranks():
    run_multiproc(_ulysses_roundtrip, 2)


def test_ulysses_shard_swap_4ranks():
    run_multiproc(_ulysses_roundtrip, 4)


def _pipeline_two_stage(rank, ws):
    """2-stage pipeline fwd+bwd must match the dense 2-layer reference."""
    from mpi4jax_amd.parallel import (send_activation, recv_activation,
                                      backward_send)

    torch.manual_seed(3)
    l0 = torch.nn.Linear(6, 5).double()
    l1 = torch.nn.Linear(5, 2).double()
    x = torch.randn(4, 6, dtype=torch.float64, requires_grad=True)

    # dense reference (identical weights on both ranks via shared seed)
    ref = l1(torch.tanh(l0(x)))
    ref.sum().backward()
    ref_l0_wgrad = l0.weight.grad.clone()
    ref_l1_wgrad = l1.weight.grad.clone()
    l0.weight.grad = l1.weight.grad = x.grad = None

    comm = m.get_world()
    if rank == 0:
        h = torch.tanh(l0(x))
        out = send_activation(h, dest=1, comm=comm)
        backward_send(out)  # grad arrives from stage 1
        assert torch.allclose(l0.weight.grad, ref_l0_wgrad, atol=1e-12)
    else:
        a = recv_activation(torch.empty(4, 5, dtype=torch.float64),
                            source=0, comm=comm)
        loss = l1(a).sum()
        loss.backward()  # routes grad(a) back to stage 0
        assert torch.allclose(l1.weight.grad, ref_l1_wgrad, atol=1e-12)


def test_pipeline_two_stage():
    run_multiproc(_pipeline_two_stage, 2)
