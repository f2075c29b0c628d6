"""bench.py driver-contract tests (CPU): JSON line shape + torchrun ws=2."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_KEYS = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
                 "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                 "dtype", "data", "config"}


def _last_json_line(out):
    for line in reversed(out.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{out}")


def test_bench_single_process_cpu():
    r = subprocess.run(
        [sys.executable, "bench.py", "--device", "cpu", "--steps", "2",
         "--warmup", "1", "--layers", "1,1,1,1"],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    j = _last_json_line(r.stdout)
    assert REQUIRED_KEYS.issubset(j.keys())
    assert j["n_gpus"] == 1 and j["steps"] == 2 and j["value"] > 0
    assert j["data"] == "synthetic" and j["scaling"] == "weak"
    assert j["config"]["parallelism"] == "dp1"


def test_bench_torchrun_ws8_cpu():
    """The 8-GPU scaling-run shape (dp8) end to end on CPU/gloo: 8 ranks,
    bucketed DP, max-over-ranks timing, one JSON line from rank 0."""
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", "29713", "bench.py", "--gpus", "8", "--steps", "1",
         "--warmup", "0", "--device", "cpu", "--layers", "1,1,1,1"],
        cwd=REPO, capture_output=True, text=True, timeout=1200, env=env)
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
    j = _last_json_line(r.stdout)
    assert j["n_gpus"] == 8
    assert j["config"]["parallelism"] == "dp8"


def test_bench_torchrun_ws2_cpu():
    """The driver launches bench via torch.distributed.run for N>1; verify the
    whole path works (gloo on CPU, 127.0.0.1 rendezvous)."""
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29711", "bench.py", "--gpus", "2", "--steps", "2",
         "--warmup", "1", "--device", "cpu", "--layers", "1,1,1,1"],
        cwd=REPO, capture_output=True, text=True, timeout=900, env=env)
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
    j = _last_json_line(r.stdout)
    assert j["n_gpus"] == 2
    assert j["config"]["parallelism"] == "dp2"
    assert j["config"]["global_batch"] == 2 * j["config"]["per_domain_batch"] * 3
