"""Cross-build comparison: fused trajectory under the normal build vs the
-ffp-contract=off build.  Saves/loads via files because the extension
choice is fixed at first import.  Prints per-field max |fma - nofma|.
"""

import os
import subprocess
import sys
import tempfile
import textwrap

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

SCRIPT = textwrap.dedent("""
    import os, sys, torch
    sys.path.insert(0, %(repo)r)
    import mpi4jax_amd as m
    m.init()
    from mpi4jax_amd.models import ShallowWater
    torch.manual_seed(0)
    sw = ShallowWater(nx=128, ny=96, device="cuda", dtype=torch.float64)
    s = sw.initial_conditions()
    s = sw.step(s, first_step=True)
    snaps = []
    for i in range(30):
        s = sw.step(s)
        if i in (0, 4, 29):
            snaps.append({k: getattr(s, k).cpu().clone()
                          for k in ("h", "u", "v")})
    torch.cuda.synchronize()
    torch.save(snaps, %(out)r)
    print("SAVED", flush=True)
""")


def run(out, nofma):
    env = dict(os.environ)
    if nofma:
        env["MPI4JAX_AMD_SW_EXT"] = "nofma"
    else:
        env.pop("MPI4JAX_AMD_SW_EXT", None)
    r = subprocess.run(
        [sys.executable, "-c", SCRIPT % {"repo": REPO, "out": out}],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stdout + r.stderr


def main():
    import torch

    with tempfile.TemporaryDirectory() as td:
        a_p, b_p = os.path.join(td, "a.pt"), os.path.join(td, "b.pt")
        run(a_p, False)
        run(b_p, True)
        a, b = torch.load(a_p), torch.load(b_p)
    for snap_i, (sa, sb) in zip((1, 5, 30), zip(a, b)):
        for k in ("h", "u", "v"):
            d = (sa[k] - sb[k]).abs()
            n = (sa[k] != sb[k]).sum().item()
            print(f"step {snap_i:>2} {k}: max|fma-nofma|={d.max().item():.3e}"
                  f"  differing cells={n}/{sa[k].numel()}", flush=True)


if __name__ == "__main__":
    main()
