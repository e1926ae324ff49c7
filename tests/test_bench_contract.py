"""bench.py contract tests — the driver depends on this exact interface.

Runs bench.py as the driver does: single process, and under
torch.distributed.run with 2 CPU ranks (gloo; nccl==RCCL takes this path on
GPU boxes). Verifies the one-line JSON contract fields.
"""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(ROOT, "bench.py")


def last_json_line(out: str) -> dict:
    for line in reversed(out.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{out}")


@pytest.mark.timeout(300)
def test_bench_single_process():
    env = dict(os.environ, AIRTC_BENCH_FAMILY="tiny")
    r = subprocess.run(
        [sys.executable, BENCH, "--steps", "2", "--warmup", "1",
         "--width", "64", "--latency-frames", "2"],
        capture_output=True, text=True, timeout=280, cwd=ROOT, env=env,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    d = last_json_line(r.stdout)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config", "p50_glass_to_glass_ms"):
        assert key in d, f"missing {key}"
    assert d["n_gpus"] == 1 and d["steps"] == 2
    assert d["scaling"] == "weak" and d["higher_is_better"] is True
    assert d["value"] > 0 and d["ms_per_step"] > 0


def _free_port():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.mark.timeout(600)
def test_bench_torchrun_2rank_gloo():
    env = dict(os.environ, AIRTC_BENCH_FAMILY="tiny")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()), BENCH,
         "--gpus", "2", "--steps", "2", "--warmup", "1",
         "--width", "64", "--latency-frames", "2"],
        capture_output=True, text=True, timeout=580, cwd=ROOT, env=env,
    )
    assert r.returncode == 0, (r.stdout[-1500:], r.stderr[-1500:])
    d = last_json_line(r.stdout)
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "frame-level dp2"
    # aggregate over ranks: 2 ranks x 2 steps / max-time
    assert d["value"] > 0
