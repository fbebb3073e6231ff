"""Diagnose the f64 fused-vs-eager divergence: which step, which field,
and is it deterministic?  Run on a GPU box:

    python tools/fma_diag.py            # current extension build
    MPI4JAX_AMD_SW_EXT=nofma python tools/fma_diag.py

Prints per-step, per-field max |fused - eager| for the first steps, plus
a determinism check (two fused runs must be bitwise identical — rules out
races in the fused kernels/exchange).
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

import mpi4jax_amd as m  # noqa: E402
from mpi4jax_amd.models import ShallowWater  # noqa: E402


def run_traj(fused, steps, seed=0):
    torch.manual_seed(seed)
    sw = ShallowWater(nx=128, ny=96, device="cuda", dtype=torch.float64,
                      fused=fused)
    s = sw.initial_conditions()
    s = sw.step(s, first_step=True)
    out = [{k: getattr(s, k).clone() for k in ("h", "u", "v")}]
    for _ in range(steps):
        s = sw.step(s)
        out.append({k: getattr(s, k).clone() for k in ("h", "u", "v")})
    torch.cuda.synchronize()
    return out


def main():
    m.init()
    from mpi4jax_amd._backend import rccl

    print("extension:", rccl.ext().__file__, flush=True)
    steps = int(os.environ.get("FMA_DIAG_STEPS", "8"))

    tf = run_traj(True, steps)
    te = run_traj(False, steps)
    for i, (a, b) in enumerate(zip(tf, te)):
        errs = {k: (a[k] - b[k]).abs().max().item() for k in a}
        label = "first_step" if i == 0 else f"step {i}"
        print(f"{label:>12}: " +
              "  ".join(f"{k}={v:.3e}" for k, v in errs.items()),
              flush=True)

    # determinism: same build, same seed, twice — must be bitwise equal
    tf2 = run_traj(True, steps)
    det = all(torch.equal(a[k], b[k])
              for a, b in zip(tf, tf2) for k in a)
    print("fused determinism (2 runs bitwise equal):", det, flush=True)
    te2 = run_traj(False, steps)
    det_e = all(torch.equal(a[k], b[k])
                for a, b in zip(te, te2) for k in a)
    print("eager determinism (2 runs bitwise equal):", det_e, flush=True)


if __name__ == "__main__":
    main()
