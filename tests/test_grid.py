import pytest
import torch

import mpi4jax_amd as m
from mpi4jax_amd.parallel.grid import CartesianGrid, default_dims


def test_default_dims():
    assert default_dims(1) == (1, 1)
    assert default_dims(2) == (2, 1)
    assert default_dims(4) == (2, 2)
    assert default_dims(8) == (2, 4)
    assert default_dims(16) == (2, 8)


def test_neighbors_single_periodic_x():
    g = CartesianGrid(m.COMM_WORLD, dims=(1, 1), periodic=(False, True))
    assert g.neighbor("north") is None
    assert g.neighbor("south") is None
    # periodic x with one column: neighbor is itself
    assert g.neighbor("east") == 0
    assert g.neighbor("west") == 0


def test_halo_exchange_self_periodic():
    # 1-rank periodic-x halo exchange: east halo <- west interior col etc.
    g = CartesianGrid(m.COMM_WORLD, dims=(1, 1), periodic=(False, True))
    a = torch.arange(20, dtype=torch.float32).reshape(4, 5)
    out = g.halo_exchange(a)
    assert torch.equal(out[:, 0], a[:, -2])
    assert torch.equal(out[:, -1], a[:, 1])
    # north/south halos untouched (closed)
    assert torch.equal(out[0, 1:-1], a[0, 1:-1])
    # input untouched
    assert torch.equal(a, torch.arange(20.).reshape(4, 5))


def test_bad_grid():
    with pytest.raises(ValueError):
        CartesianGrid(m.COMM_WORLD, dims=(2, 2))


def test_halo_exchange_rejects_degenerate():
    g = CartesianGrid(m.COMM_WORLD, dims=(1, 1), periodic=(False, True))
    with pytest.raises(ValueError, match="3x3"):
        g.halo_exchange(torch.zeros(2, 5))
    with pytest.raises(ValueError, match="3x3"):
        g.halo_exchange(torch.zeros(4))
