"""CLI entry points (python -m olearning_sim_amd ...) — in-process
one-shot paths: submit -> status -> queue over a temp data dir.
Reference parity: the entry scripts absent from the open-source drop
(its README's Dockerfile entrypoints)."""

import json
import os
import subprocess
import sys

from tests.test_manager import task_json


def _run(tmp, *args, inp=None):
    env = dict(os.environ)
    env.setdefault("OLSIM_FORCE_CPU", "1")
    return subprocess.run(
        [sys.executable, "-m", "olearning_sim_amd",
         "--data-dir", str(tmp / "data"), *args],
        capture_output=True, text=True, timeout=180, env=env,
        cwd="/root/repo")


def test_cli_submit_status_queue(tmp_path):
    tj = tmp_path / "task.json"
    tj.write_text(task_json(task_id="t_cli", clients=4, dynamic=1))

    out = _run(tmp_path, "submit", str(tj))
    assert out.returncode == 0, out.stderr
    line = json.loads(out.stdout.strip().splitlines()[-1])
    assert line["is_success"] is True, line

    out = _run(tmp_path, "status", "t_cli")
    assert out.returncode == 0, out.stderr
    st = json.loads(out.stdout.strip().splitlines()[-1])
    assert "task_status" in st

    out = _run(tmp_path, "queue")
    assert out.returncode == 0, out.stderr

    out = _run(tmp_path, "stop", "t_cli")
    assert out.returncode == 0, out.stderr
