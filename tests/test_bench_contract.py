"""bench.py driver-contract tests: the round-end driver runs
`python bench.py --gpus N --steps K --warmup W` (N>1 via
torch.distributed.run) and parses ONE JSON line from rank 0 — verify
both launch modes on CPU and the schema of the line."""

import json
import os
import subprocess
import sys

from conftest import free_port

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(ROOT, "bench.py")

REQUIRED_KEYS = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
                 "ms_per_step", "higher_is_better", "scaling",
                 "vs_baseline", "dtype", "data", "config"}


def _last_json_line(text: str) -> dict:
    for line in reversed(text.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{text[-2000:]}")


def test_bench_single_process(tmp_path):
    out = subprocess.run(
        [sys.executable, BENCH, "--steps", "2", "--warmup", "1"],
        cwd=ROOT, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    rec = _last_json_line(out.stdout)
    assert REQUIRED_KEYS <= set(rec)
    assert rec["n_gpus"] == 1
    assert rec["value"] > 0
    assert rec["config"]["parallelism"] == "dp1"


def test_bench_torchrun_two_ranks(tmp_path):
    """The exact multi-GPU launch shape the driver uses, on CPU/gloo."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()), BENCH, "--gpus", "2", "--steps", "2",
         "--warmup", "1"],
        cwd=ROOT, capture_output=True, text=True, timeout=600, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    rec = _last_json_line(out.stdout)
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["value"] > 0
