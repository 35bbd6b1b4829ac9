"""bench.py driver contract: the round-end driver runs
`python bench.py --gpus N --steps K --warmup W` (N>1 via
torch.distributed.run) and parses ONE JSON line from rank 0. These tests
exercise that exact surface on CPU with a tiny model so a bench.py
regression never reaches the driver."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(ROOT, "bench.py")

TINY = [
    "--model", "tiny-llama", "--batch", "8", "--prompt-len", "24",
    "--gen-tokens", "6", "--steps", "2", "--warmup", "1",
]


def run_bench(extra, env=None):
    e = dict(os.environ)
    if env:
        e.update(env)
    out = subprocess.run(
        [sys.executable, BENCH] + TINY + extra,
        capture_output=True, text=True, timeout=420, cwd=ROOT, env=e,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    return json.loads(line)


def check_schema(d, metric):
    assert d["metric"] == metric
    for key in ("value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
                "higher_is_better", "scaling", "vs_baseline", "dtype",
                "data", "config"):
        assert key in d, key
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["config"]["p50_ttft_ms"] is not None
    assert d["value"] > 0


def test_default_mode_is_fixed_qps():
    d = run_bench([])
    check_schema(d, "req/sec @ fixed QPS")
    assert d["config"]["mode"] == "open-loop-poisson"
    assert d["config"]["qps_offered_per_gpu"] > 0


def test_wave_mode():
    d = run_bench(["--mode", "wave"])
    check_schema(d, "req/sec")
    assert d["config"]["mode"] == "closed-loop-waves"


@pytest.mark.timeout(420)
def test_two_rank_launch_matches_driver_shape():
    env = dict(os.environ)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29617", BENCH, "--gpus", "2"] + TINY,
        capture_output=True, text=True, timeout=400, cwd=ROOT, env=env,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, "exactly one JSON line (rank 0)"
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 16  # whole-job aggregate
