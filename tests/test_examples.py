"""Examples-as-tests (the reference runs its demo in CI the same way,
tests/test_examples.py:20-24)."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
ENV = {k: v for k, v in os.environ.items()
       if not k.startswith(("RANK", "WORLD_SIZE", "MASTER_", "LOCAL_"))}
ENV["PYTHONPATH"] = REPO


def test_shallow_water_example_single():
    res = subprocess.run(
        [sys.executable, "examples/shallow_water.py", "--nx", "72",
         "--ny", "36", "--days", "0.02"],
        capture_output=True, text=True, timeout=300, env=ENV, cwd=REPO,
    )
    assert res.returncode == 0, res.stderr
    assert "Solution took" in res.stdout


def test_shallow_water_example_two_ranks():
    res = subprocess.run(
        [sys.executable, "-m", "mpi4jax_amd.run", "-n", "2",
         "examples/shallow_water.py", "--nx", "72", "--ny", "36",
         "--days", "0.01"],
        capture_output=True, text=True, timeout=300, env=ENV, cwd=REPO,
    )
    assert res.returncode == 0, res.stderr + res.stdout
    assert "Solution took" in res.stdout


def test_distributed_cg_example():
    res = subprocess.run(
        [sys.executable, "examples/distributed_cg.py", "--n", "128"],
        capture_output=True, text=True, timeout=300, env=ENV, cwd=REPO,
    )
    assert res.returncode == 0, res.stderr
    assert "iterations" in res.stdout


def test_tensor_parallel_example_two_ranks():
    res = subprocess.run(
        [sys.executable, "-m", "mpi4jax_amd.run", "-n", "2",
         "examples/tensor_parallel_mlp.py", "--dim", "64", "--steps", "6"],
        capture_output=True, text=True, timeout=300, env=ENV, cwd=REPO,
    )
    assert res.returncode == 0, res.stderr + res.stdout
    assert "OK: 6 TP steps over 2 rank(s)" in res.stdout


def test_master_worker_example_three_ranks():
    """Dynamic task farm: ANY_SOURCE + ANY_TAG + Status-driven work
    hand-out (the envelope plane's flagship CPU use)."""
    res = subprocess.run(
        [sys.executable, "-m", "mpi4jax_amd.run", "-n", "3",
         "examples/master_worker.py", "--items", "12"],
        capture_output=True, text=True, timeout=300, env=ENV, cwd=REPO,
    )
    assert res.returncode == 0, res.stderr + res.stdout
    assert "master: 12/12 items verified OK" in res.stdout
