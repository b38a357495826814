"""bench.py contract tests on CPU: single-process JSON line and the
torchrun multi-process path the driver uses for SCALE runs."""

import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _last_json(out):
    for line in reversed(out.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in:\n{out}")


def test_bench_single_process_json():
    p = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "1",
         "--batch", "2", "--model", "mobilenet_v3_small"],
        cwd=ROOT, capture_output=True, text=True, timeout=600)
    assert p.returncode == 0, p.stderr[-2000:]
    d = _last_json(p.stdout)
    assert d["n_gpus"] == 1 and d["unit"] == "img/s"
    assert d["higher_is_better"] is True and d["data"] == "synthetic"
    assert d["config"]["parallelism"] == "dp1"
    assert d["value"] > 0 and d["ms_per_step"] > 0


def test_bench_torchrun_two_ranks():
    env = dict(os.environ, GLOO_SOCKET_IFNAME="lo")
    p = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", "bench.py", "--gpus", "2",
         "--steps", "1", "--warmup", "1", "--batch", "2",
         "--model", "mobilenet_v3_small"],
        cwd=ROOT, capture_output=True, text=True, timeout=600, env=env)
    assert p.returncode == 0, p.stderr[-2000:]
    d = _last_json(p.stdout)
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 4  # 2 per rank, whole-job
