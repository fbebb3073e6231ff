import os, sys
sys.path.insert(0, "/root/repo")
os.environ["MPI4JAX_AMD_SW_GRAPH"] = "0"
import torch
from mpi4jax_amd.models import ShallowWater
sw = ShallowWater(nx=3600, ny=1800, device="cuda")
s = sw.initial_conditions()
s = sw.step(s, first_step=True)
for _ in range(10):
    s = sw.step(s)
torch.cuda.synchronize()
print("probe done")
