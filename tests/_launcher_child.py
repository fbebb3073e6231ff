"""Child script for the launcher test (2 ranks over gloo)."""

import torch

import mpi4jax_amd as m

m.init()
comm = m.get_world()
x = torch.tensor([float(comm.rank + 1)])
y = m.allreduce(x, m.SUM)
if comm.rank == 0:
    print(f"SUM={y.item()}", flush=True)
m.finalize()
