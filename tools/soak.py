"""Long-run stability soak for the shallow-water benchmark config.

Reproduces the stability evidence in profiles/README.md: N model days at
(3600, 1800) f32 on one GPU (hipGraph multistep), checking that the mean
surface height stays at the 100 m mean depth and every field stays
finite.

Run:  python tools/soak.py [--days D] [--nx NX] [--ny NY]
(~0.5 s of wall per model day at the current kernel generation)
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402

from mpi4jax_amd.models import ShallowWater  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--days", type=float, default=1.0)
    p.add_argument("--nx", type=int, default=3600)
    p.add_argument("--ny", type=int, default=1800)
    args = p.parse_args()

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    sw = ShallowWater(nx=args.nx, ny=args.ny, device=dev)
    t0 = time.perf_counter()
    state, steps, wall = sw.solve(args.days * 86_400.0,
                                  num_multisteps=500 if sw.fused else 50)
    total = time.perf_counter() - t0

    h_int = state.h[1:-1, 1:-1]
    report = {
        "days": args.days,
        "steps": steps,
        "ms_per_step": round(wall / steps * 1e3, 4),
        "solve_wall_s": round(wall, 2),
        "total_wall_s": round(total, 2),
        "mean_height_m": round(h_int.mean().item(), 4),
        "height_range_m": [round(h_int.min().item(), 2),
                           round(h_int.max().item(), 2)],
        "max_speed_ms": round(max(state.u.abs().max().item(),
                                  state.v.abs().max().item()), 2),
        "all_finite": bool(torch.isfinite(state.h).all()
                           and torch.isfinite(state.u).all()
                           and torch.isfinite(state.v).all()),
        "device": dev,
    }
    print(json.dumps(report))
    assert report["all_finite"], "model diverged"
    assert abs(report["mean_height_m"] - 100.0) < 0.01, "mass not conserved"


if __name__ == "__main__":
    main()
