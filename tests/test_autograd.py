"""Autodiff rules (single-process leg).

Reference semantics: allreduce(SUM) VJP is the identity and its JVP is an
allreduce of the tangent (``allreduce.py:138-159``); sendrecv's VJP swaps
source/dest (``sendrecv.py:278-293``).
"""

import pytest
import torch
import torch.autograd.forward_ad as fwd_ad

import mpi4jax_amd as m


def test_allreduce_backward_identity():
    x = torch.randn(4, requires_grad=True)
    y = m.allreduce(x, m.SUM)
    g = torch.randn(4)
    y.backward(g)
    assert torch.equal(x.grad, g)


def test_allreduce_grad_of_sum():
    x = torch.randn(5, requires_grad=True)
    loss = m.allreduce(x, m.SUM).sum()
    loss.backward()
    assert torch.equal(x.grad, torch.ones(5))


def test_allreduce_jvp():
    x = torch.randn(4)
    t = torch.randn(4)
    with fwd_ad.dual_level():
        xd = fwd_ad.make_dual(x, t)
        yd = m.allreduce(xd, m.SUM)
        y, yt = fwd_ad.unpack_dual(yd)
    assert torch.allclose(y, x)
    assert torch.allclose(yt, t)


def test_allreduce_nonsum_grad_raises():
    x = torch.randn(4, requires_grad=True)
    y = m.allreduce(x, m.MAX)
    with pytest.raises(RuntimeError, match="only differentiable"):
        y.sum().backward()


def test_sendrecv_backward_self():
    x = torch.randn(4, requires_grad=True)
    y = m.sendrecv(x, x.detach(), source=0, dest=0)
    g = torch.randn(4)
    y.backward(g)
    assert torch.equal(x.grad, g)


def test_chained_ops_grad():
    # distributed matvec pattern (reference test_allreduce_matvec.py)
    A = torch.randn(3, 3)
    x = torch.randn(3, requires_grad=True)
    y = m.allreduce(A @ x, m.SUM)
    y.sum().backward()
    assert torch.allclose(x.grad, A.t() @ torch.ones(3))


def test_sendrecv_jvp_self():
    """Forward-mode tangent flows along the same edge; shapes can differ
    between send and recv buffers."""
    x = torch.randn(2, 3)
    t = torch.randn(2, 3)
    tmpl = torch.empty(6)  # different recv shape (reshaped message)
    with fwd_ad.dual_level():
        xd = fwd_ad.make_dual(x, t)
        yd = m.sendrecv(xd, tmpl, source=0, dest=0)
        y, yt = fwd_ad.unpack_dual(yd)
    assert torch.equal(y, x.reshape(6))
    assert torch.equal(yt, t.reshape(6))


def test_compose_with_custom_autograd_function():
    """Collectives inside user custom Functions (analog of the reference's
    custom_vjp interplay tests, test_allreduce.py:226-322)."""

    class Scale2(torch.autograd.Function):
        @staticmethod
        def forward(ctx, x):
            return m.allreduce(x.detach() * 2, m.SUM)

        @staticmethod
        def backward(ctx, g):
            return m.allreduce(g, m.SUM) * 2

    x = torch.randn(4, requires_grad=True)
    y = Scale2.apply(x)
    assert torch.allclose(y, 2 * x)
    y.sum().backward()
    assert torch.allclose(x.grad, 2 * torch.ones(4))


def test_compose_with_checkpointing():
    """allreduce under torch.utils.checkpoint: the recompute path re-runs
    the collective symmetrically on every rank."""
    from torch.utils.checkpoint import checkpoint

    def block(x):
        return torch.tanh(m.allreduce(x, m.SUM)) * 3

    x = torch.randn(5, dtype=torch.float64, requires_grad=True)
    y = checkpoint(block, x, use_reentrant=False)
    y.sum().backward()
    expect = 3 * (1 - torch.tanh(x.detach()) ** 2)
    assert torch.allclose(x.grad, expect, atol=1e-12)
