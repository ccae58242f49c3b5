"""bench.py driver-contract tests: single-process and 2-rank torchrun
(gloo on CPU here; the same code path is RCCL on the GPU node)."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_KEYS = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
                 "ms_per_step", "higher_is_better", "scaling",
                 "vs_baseline", "dtype", "data", "config"}


def last_json_line(stdout: str) -> dict:
    for line in reversed(stdout.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{stdout}")


@pytest.mark.timeout(300)
def test_bench_single_process_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--preset", "mlp-cpu",
         "--steps", "3", "--warmup", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-2000:]
    d = last_json_line(out.stdout)
    assert REQUIRED_KEYS.issubset(d.keys())
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["n_gpus"] == 1 and d["steps"] == 3 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["config"]["parallelism"] == "dp1"
    assert d["config"]["clients_total"] == 10


@pytest.mark.timeout(420)
def test_bench_two_rank_torchrun_contract():
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--standalone",
         "--local-addr", "127.0.0.1", "--nproc-per-node", "2",
         "bench.py", "--gpus", "2", "--preset", "mlp-cpu",
         "--steps", "2", "--warmup", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=360, env=env)
    assert out.returncode == 0, out.stderr[-3000:]
    d = last_json_line(out.stdout)
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    # whole-job aggregate: both ranks' client shards count
    assert d["config"]["clients_total"] == 20
    # exactly ONE json line (rank 0 only)
    json_lines = [ln for ln in out.stdout.splitlines()
                  if ln.strip().startswith("{")]
    assert len(json_lines) == 1


def test_bench_trace_flag_writes_chrome_trace(tmp_path):
    import subprocess, sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    trace = tmp_path / "trace.json"
    out = subprocess.run(
        [sys.executable, "bench.py", "--preset", "mlp-cpu",
         "--steps", "2", "--warmup", "1", "--trace", str(trace)],
        cwd=repo, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    d = json.loads(out.stdout.strip().splitlines()[-1])
    assert d["steps"] == 2
    tr = json.loads(trace.read_text())
    assert len(tr["traceEvents"]) > 0
