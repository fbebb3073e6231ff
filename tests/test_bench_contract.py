"""Pin the driver-facing bench.py contract on CPU.

The driver runs ``python bench.py --gpus N --steps K --warmup W`` (N>1
under torch.distributed.run) and parses ONE JSON line from rank 0; it
also relies on ``--preflight`` never being broken so the first multi-GPU
contact cannot die on schedule bugs.
"""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args, timeout=420):
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    return subprocess.run(
        [sys.executable] + args, cwd=REPO, env=env, timeout=timeout,
        capture_output=True, text=True,
    )


def test_preflight_world2():
    r = _run(["-m", "torch.distributed.run", "--nnodes=1",
              "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
              "--master-port", "29591", "bench.py", "--preflight"])
    assert r.returncode == 0, r.stdout + r.stderr
    assert "PREFLIGHT_OK world_size=2" in r.stdout


def test_bench_json_contract_cpu():
    r = _run(["bench.py", "--steps", "4", "--warmup", "1",
              "--nx", "360", "--ny", "180"])
    assert r.returncode == 0, r.stdout + r.stderr
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["metric"] == "shallow_water_steps_per_sec"
    assert d["steps"] == 4 and d["warmup"] == 1
    assert d["n_gpus"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "strong"
    assert d["data"] == "synthetic"
    assert "sec_per_model_day" in d["config"]
    assert d["value"] > 0


@pytest.mark.gpu
def test_bench_comm_configs_gpu():
    """On a GPU box the driver-facing line must carry every BASELINE comm
    config, n_gpus-labeled, with the N=1 local-copy caveat."""
    r = _run(["bench.py", "--steps", "20", "--warmup", "5"], timeout=900)
    assert r.returncode == 0, r.stdout + r.stderr
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    cfg = d["config"]
    ar = cfg["allreduce_256MiB_bf16"]
    assert "error" not in ar, ar
    assert ar["n_gpus"] == 1 and "note" in ar
    bis = cfg["bisection_1GiB_bf16"]
    assert "error" not in bis, bis
    assert bis["alltoall"]["n_gpus"] == 1
    assert bis["allgather"]["n_gpus"] == 1
    ga = cfg["grad_allreduce_256MiB_bf16"]
    assert "error" not in ga, ga
    assert ga["ms"] > 0
