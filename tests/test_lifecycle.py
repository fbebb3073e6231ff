"""Process-lifecycle tests via subprocess.

Mirrors the reference's subprocess strategy (``tests/test_common.py:13-57``):
scripts run in a scrubbed environment so the framework initializes fresh;
assertions are on exit codes and stderr/stdout.
"""

import os
import subprocess
import sys
import textwrap

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_script(body, env_extra=None, timeout=120):
    env = {k: v for k, v in os.environ.items()
           if not k.startswith(("RANK", "WORLD_SIZE", "MASTER_", "LOCAL_"))}
    env["PYTHONPATH"] = REPO
    if env_extra:
        env.update(env_extra)
    return subprocess.run(
        [sys.executable, "-c", textwrap.dedent(body)],
        capture_output=True, text=True, timeout=timeout, env=env, cwd=REPO,
    )


def test_bad_rank_raises():
    # analog of the reference's MPI_Abort-on-bad-rank test
    # (test_common.py:60-88): sending to a nonexistent rank must fail
    # loudly, not hang
    res = run_script("""
        import torch, mpi4jax_amd as m
        m.init()
        m.send(torch.zeros(3), 100)
    """)
    assert res.returncode != 0
    assert "invalid dest" in res.stderr


def test_clean_exit_no_deadlock():
    # the atexit flush must let the interpreter exit after pending
    # communication (reference test_common.py:91-115)
    res = run_script("""
        import torch, mpi4jax_amd as m
        m.init()
        y = m.allreduce(torch.ones(10), m.SUM)
        assert y.sum().item() == 10
        print("DONE")
    """)
    assert res.returncode == 0, res.stderr
    assert "DONE" in res.stdout


def test_debug_env_enables_logging():
    res = run_script("""
        import torch, mpi4jax_amd as m
        m.init()
        m.allreduce(torch.zeros(3), m.SUM)
    """, env_extra={"MPI4JAX_AMD_DEBUG": "1"})
    assert res.returncode == 0, res.stderr
    assert "Allreduce" in res.stdout
    assert "done with code 0" in res.stdout


def test_no_debug_by_default():
    res = run_script("""
        import torch, mpi4jax_amd as m
        m.init()
        m.allreduce(torch.zeros(3), m.SUM)
    """)
    assert res.returncode == 0
    assert "Allreduce" not in res.stdout


def test_import_does_not_initialize():
    # unlike the reference (import runs MPI_Init), import here is inert —
    # but the first op initializes implicitly, so behavior matches
    res = run_script("""
        import mpi4jax_amd as m
        import mpi4jax_amd.parallel.comm as c
        assert c._WORLD is None, "import must not initialize"
        import torch
        m.allreduce(torch.zeros(2), m.SUM)  # implicit init
        assert c._WORLD is not None
        print("OK")
    """)
    assert res.returncode == 0, res.stderr
    assert "OK" in res.stdout


def test_version_info():
    res = run_script("""
        import mpi4jax_amd as m
        print(m.__version__)
        import mpi4jax_amd._rccl_C as ext
        info = ext.version_info()
        assert info["rccl"] > 0 and info["hip_runtime"] > 0
        print("INFO_OK")
    """)
    assert res.returncode == 0, res.stderr
    assert "INFO_OK" in res.stdout


def test_launcher_usage_error():
    res = subprocess.run(
        [sys.executable, "-m", "mpi4jax_amd.run"],
        capture_output=True, text=True, timeout=60,
        env={**os.environ, "PYTHONPATH": REPO}, cwd=REPO,
    )
    assert res.returncode == 2
    assert "usage" in res.stderr


def test_launcher_runs_script():
    script = os.path.join(REPO, "tests", "_launcher_child.py")
    res = subprocess.run(
        [sys.executable, "-m", "mpi4jax_amd.run", "-n", "2", script],
        capture_output=True, text=True, timeout=180,
        env={**os.environ, "PYTHONPATH": REPO}, cwd=REPO,
    )
    assert res.returncode == 0, res.stderr + res.stdout
    assert "SUM=3.0" in res.stdout


def test_run_launcher_out_of_tree_script(tmp_path):
    """python -m mpi4jax_amd.run must make the source-tree package
    importable for user scripts living anywhere (the child's sys.path[0]
    is the script's own directory)."""
    import subprocess
    import sys

    script = tmp_path / "user_script.py"
    # per-rank result files: rank stdout gets interleaved by the launcher
    script.write_text(
        "import pathlib, torch, mpi4jax_amd as m\n"
        "m.init()\n"
        "y = m.allreduce(torch.ones(2), m.SUM)\n"
        "r = m.get_world().rank\n"
        f"pathlib.Path(r'{tmp_path}', f'out_{{r}}.txt')"
        ".write_text(str(y.sum().item()))\n"
    )
    res = subprocess.run(
        [sys.executable, "-m", "mpi4jax_amd.run", "-n", "2", str(script)],
        capture_output=True, text=True, timeout=240,
    )
    assert res.returncode == 0, res.stdout + res.stderr
    for r in (0, 1):
        assert (tmp_path / f"out_{r}.txt").read_text() == "4.0", res.stdout


def test_jit_ops_importable_as_submodule():
    import importlib

    mod = importlib.import_module("mpi4jax_amd.jit_ops")
    import mpi4jax_amd as m

    assert mod is m.jit_ops
