import os, torch
import mpi4jax_amd as m

m.init(device=0)  # both ranks on cuda:0
comm = m.get_world()
r, ws = comm.rank, comm.size
x = torch.full((8,), float(r + 1), device="cuda")
y = m.allreduce(x, m.SUM)
torch.cuda.synchronize()
expect = sum(range(1, ws + 1))
assert y[0].item() == expect, (r, y[0].item())
other = (r + 1) % ws
z = m.sendrecv(x, x, source=other, dest=other)
torch.cuda.synchronize()
assert z[0].item() == other + 1, (r, z[0].item())
if r == 0:
    print("TWO_RANK_RCCL_OK", flush=True)
