"""Dry-run of the driver's EXACT multi-GPU launch line on CPU
(DPRK_BENCH_SIM=1): catches control-plane bugs (uid broadcast, count
matrix, recv bookkeeping, arg plumbing) before the one-shot round-end
8-GPU run.  The sim engine (dpark_amd/testing.py) is test infrastructure
only — output is marked SIMULATED."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
@pytest.mark.parametrize("op,n", [("reduce", 2), ("sort", 2), ("reduce", 4)])
def test_torchrun_launch_line(op, n):
    pytest.importorskip("torch")
    env = dict(os.environ)
    env["DPRK_BENCH_SIM"] = "1"
    port = 29600 + n + (7 if op == "sort" else 0)
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           f"--nproc-per-node={n}", "--master-addr", "127.0.0.1",
           "--master-port", str(port), "bench.py", "--gpus", str(n),
           "--steps", "2", "--warmup", "1", "--op", op,
           "--n-pairs", "200000", "--skip-cpu-baseline"]
    out = subprocess.run(cmd, cwd=ROOT, env=env, capture_output=True,
                         text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-3000:]
    line = [ln for ln in out.stdout.splitlines() if ln.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == n
    assert d["data"] == "SIMULATED-no-gpu"
    assert d["scaling"] == "weak"
    assert d["config"]["parallelism"] == f"dp{n}"


@pytest.mark.timeout(300)
def test_plain_gpus_flag_self_bootstraps():
    """`python bench.py --gpus N` WITHOUT torchrun must
    bootstrap N ranks itself (exec torch.distributed.run), not silently
    measure N=1."""
    pytest.importorskip("torch")
    env = dict(os.environ)
    env["DPRK_BENCH_SIM"] = "1"
    env.pop("WORLD_SIZE", None)
    cmd = [sys.executable, "bench.py", "--gpus", "2", "--steps", "2",
           "--warmup", "1", "--op", "reduce", "--n-pairs", "200000",
           "--skip-cpu-baseline"]
    out = subprocess.run(cmd, cwd=ROOT, env=env, capture_output=True,
                         text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-3000:]
    d = json.loads([ln for ln in out.stdout.splitlines()
                    if ln.startswith("{")][-1])
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"


@pytest.mark.timeout(120)
def test_single_rank_sim():
    env = dict(os.environ)
    env["DPRK_BENCH_SIM"] = "1"
    out = subprocess.run([sys.executable, "bench.py", "--steps", "2",
                          "--warmup", "1", "--n-pairs", "100000",
                          "--skip-cpu-baseline"],
                         cwd=ROOT, env=env, capture_output=True, text=True,
                         timeout=110)
    assert out.returncode == 0, out.stderr[-2000:]
    d = json.loads([ln for ln in out.stdout.splitlines()
                    if ln.startswith("{")][-1])
    assert d["n_gpus"] == 1 and d["data"] == "SIMULATED-no-gpu"
