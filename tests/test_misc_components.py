"""File repo, operator ABCs, deviceflow strategy validation, checkpoint
naming, logger table, node manager worker launch."""

import json
import os

import pytest
import torch

from olearning_sim_amd.utils.file_repo import LocalFileRepo
from olearning_sim_amd.deviceflow.validate import ValidateStrategy
from olearning_sim_amd.engine.checkpoint import (checkpoint_name,
                                                 save_checkpoint,
                                                 load_checkpoint)


def test_file_repo_cycle(tmp_path):
    repo = LocalFileRepo(root=str(tmp_path / "store"))
    src = tmp_path / "a.bin"
    src.write_bytes(b"hello")
    assert repo.upload_file(str(src), "bucket1", "dir/a.bin")
    assert repo.bucket_exists("bucket1")
    assert repo.exists("bucket1", "dir/a.bin")
    assert repo.list_files("bucket1") == ["dir/a.bin"]
    dst = tmp_path / "out.bin"
    assert repo.download_file("bucket1", "dir/a.bin", str(dst))
    assert dst.read_bytes() == b"hello"
    # download_payload removes the object (reference semantics)
    dst2 = tmp_path / "out2.bin"
    assert repo.download_payload("bucket1", "dir/a.bin", str(dst2))
    assert not repo.exists("bucket1", "dir/a.bin")
    assert not repo.download_file("bucket1", "dir/a.bin", str(dst))


def test_operator_abc_params_schema():
    from olearning_sim_amd.task.operator_base import OperatorABC
    class MyOp(OperatorABC):
        def construct_run_params(self): return {}
        def construct_run_script(self): return "train.py"
        def run(self): return 0
    op = MyOp()
    payload = {"task_id": "t", "current_round": 2,
               "data": {"name": "d0", "task_type": "classification"},
               "operator": {"name": "train", "operator_params": "{}"},
               "actor_simulation_num": 4, "params": {}}
    got = op.get_params(["--params", json.dumps(payload)])
    assert got["task_id"] == "t" and got["current_round"] == 2
    assert got["data"]["name"] == "d0"


@pytest.mark.parametrize("spec,ok", [
    ({"real_time_dispatch": {"use_strategy": True,
                             "dispatch_batch_sizes": [4]}}, True),
    ({"real_time_dispatch": {"use_strategy": True},
      "flow_dispatch": {"use_strategy": True}}, False),   # both
    ({}, False),                                           # neither
    ({"flow_dispatch": {"use_strategy": True, "total_dispatch_amount": 0,
                        "specific_timing": {"use": True}}}, False),
    ({"flow_dispatch": {"use_strategy": True, "total_dispatch_amount": 10,
                        "specific_timing": {"use": True,
                                            "timings": [0, 5],
                                            "amounts": [5, 5]}}}, True),
    ({"flow_dispatch": {"use_strategy": True, "total_dispatch_amount": 10,
                        "specific_timing": {"use": True,
                                            "timings": [5, 0],
                                            "amounts": [5, 5]}}}, False),
    ({"flow_dispatch": {"use_strategy": True, "total_dispatch_amount": 10,
                        "specific_interval": {
                            "use": True, "intervals": [[0, 5], [3, 8]],
                            "dispatch_rules": {
                                "domains": [[0, 1], [0, 1]],
                                "functions": ["1", "1"]}}}}, False),  # overlap
    ({"flow_dispatch": {"use_strategy": True, "total_dispatch_amount": 10,
                        "specific_interval": {
                            "use": True, "intervals": [[0, 5]],
                            "dispatch_rules": {"domains": [[0, 6.28]],
                                               "functions": ["math.sin(t)+1"]},
                            "drop_simulation": {"drop_probability": [0.5],
                                                "drop_amounts": [1]}}}},
     False),                                               # two drop keys
    ({"flow_dispatch": {"use_strategy": True, "total_dispatch_amount": 10,
                        "specific_interval": {
                            "use": True, "intervals": [[0, 5]],
                            "dispatch_rules": {"domains": [[0, 6.28]],
                                               "functions": ["math.sin(t)+1"]},
                            "drop_simulation": {"drop_probability": [1.5]}}}},
     False),                                               # p out of range
])
def test_strategy_validation(spec, ok):
    v = ValidateStrategy()
    assert v.check(json.dumps(spec)) == ok, v.last_error


def test_checkpoint_naming_and_roundtrip(tmp_path):
    assert checkpoint_name("taskX", 3) == \
        "taskX_3_result_model.safetensors"
    assert checkpoint_name("t", 1, "{task_id}_{current_round}_result_model.mnn") \
        == "t_1_result_model.mnn"
    state = {"w": torch.randn(4, 3), "b": torch.zeros(4)}
    path = save_checkpoint(str(tmp_path), "taskX", 2, state)
    assert os.path.basename(path) == "taskX_2_result_model.safetensors"
    back = load_checkpoint(str(tmp_path), "taskX", 2)
    torch.testing.assert_close(back["w"], state["w"])
    assert load_checkpoint(str(tmp_path), "taskX", 9) is None


def test_engine_checkpoints_every_round(tmp_path):
    from olearning_sim_amd.engine import EngineJob, LogicalEngine
    job = EngineJob(task_id="ck", model_name="mlp",
                    model_kwargs={"in_features": 16, "hidden": 8,
                                  "num_classes": 4},
                    clients=4, rounds=3, local_steps=1, batch_size=2,
                    lr=0.1, num_classes=4, checkpoint_dir=str(tmp_path),
                    save_every_round=True, seed=0)
    out = LogicalEngine(job).run()
    assert out["rounds"] == 3
    for r in range(3):
        assert os.path.exists(tmp_path / f"ck_{r}_result_model.safetensors")
    # resume: load round-1 weights into a fresh engine
    eng2 = LogicalEngine(job)
    sd = load_checkpoint(str(tmp_path), "ck", 1)
    eng2.master.load_state_dict(sd)
    for k, v in sd.items():   # safetensors reorders keys; compare by name
        torch.testing.assert_close(eng2.master.views[k], v)


def test_logger_writes_table(tmp_path):
    from olearning_sim_amd.utils.logging import Logger
    lg = Logger(log_dir=str(tmp_path), db_path=str(tmp_path / "log.sqlite"))
    lg.info("t1", "TaskMgr", "test", "hello world")
    lg.error("t1", "TaskMgr", "test", "bad thing")
    rows = lg._table.get_rows_where({"task_id": "t1"})
    assert len(rows) == 2
    assert {r["log_type"] for r in rows} == {"info", "error"}


def test_operatorflow_gates(tmp_path):
    import threading, time
    from olearning_sim_amd.engine.operatorflow import OperatorFlow, GateTimeout

    # empty strategy = immediate
    OperatorFlow("t").start(0)
    OperatorFlow("t").stop(0)

    # selection-service gate: stop waits until the round advances
    state = {"round": 0}
    flow = OperatorFlow("t", stop_strategy="waiting_for_global_aggregation",
                        wait_interval=0.02,
                        selection_round_fn=lambda: state["round"])
    t = threading.Thread(target=lambda: (time.sleep(0.1),
                                         state.update(round=1)))
    t.start()
    flow.stop(0)          # returns once service round > 0
    t.join()
    assert state["round"] == 1

    # flag-file handshake
    flow = OperatorFlow("t", stop_strategy="sample_and_aggregation",
                        wait_interval=0.02, work_dir=str(tmp_path))
    def aggregator():
        while not (tmp_path / "simulation_finished.txt").exists():
            time.sleep(0.01)
        (tmp_path / "aggregation_finished.txt").write_text("done")
    t = threading.Thread(target=aggregator)
    t.start()
    flow.stop(0)
    t.join()
    assert not (tmp_path / "aggregation_finished.txt").exists()  # consumed

    # timeout honoured
    flow = OperatorFlow("t", stop_strategy="waiting_for_global_aggregation",
                        wait_interval=0.01, total_timeout=0.05,
                        selection_round_fn=lambda: 0)
    with pytest.raises(GateTimeout):
        flow.stop(0)


def test_fragment_repo_json_and_tensor():
    from olearning_sim_amd.deviceflow.rooms import OutboundRoom, Message
    from olearning_sim_amd.utils.fragments import (JsonFragmentRepo,
                                                   TensorFragmentRepo)
    out = OutboundRoom()
    out.send(Message("t_train_0", "logical_simulation",
                     payload='{"grad_norm": 1.5}'))
    import base64
    out.send(Message("t_train_0", "logical_simulation",
                     payload=base64.b64encode(b'{"grad_norm": 2.5}')))
    frags = list(JsonFragmentRepo(out).fragments())
    assert [f["payload"]["grad_norm"] for f in frags] == [1.5, 2.5]

    out2 = OutboundRoom()
    t = torch.ones(3)
    out2.send(Message("t_train_0", "logical_simulation", payload=t))
    frags = list(TensorFragmentRepo(out2).fragments())
    assert torch.equal(frags[0]["payload"], t)


def test_hybrid_data_splitter(tmp_path):
    from olearning_sim_amd.utils.data_split import HybridDataSplitter
    data = tmp_path / "data"
    data.mkdir()
    rows = "\n".join(f"{i},{i*2}" for i in range(100))
    (data / "train.csv").write_text("a,b\n" + rows + "\n")
    (data / "meta.txt").write_text("labels")
    sp = HybridDataSplitter(seed=3)
    nl, nd = sp.split_dir(str(data), str(tmp_path / "lg"), str(tmp_path / "dv"),
                          device_fraction=0.3)
    assert nl == 70 and nd == 30
    lg = (tmp_path / "lg" / "train.csv").read_text().splitlines()
    dv = (tmp_path / "dv" / "train.csv").read_text().splitlines()
    assert lg[0] == dv[0] == "a,b"
    assert len(lg) - 1 == 70 and len(dv) - 1 == 30
    # every row lands exactly once
    assert sorted(lg[1:] + dv[1:]) == sorted(rows.split("\n"))
    assert (tmp_path / "lg" / "meta.txt").exists()
    assert (tmp_path / "dv" / "meta.txt").exists()
    # archives round-trip
    z = sp.archive(str(tmp_path / "lg"), str(tmp_path / "lg.zip"))
    out = sp.extract(z, str(tmp_path / "back"))
    assert (tmp_path / "back" / "train.csv").exists()


def test_simulator_config_load(tmp_path):
    from olearning_sim_amd.config import SimulatorConfig
    cfg = SimulatorConfig.load(None)
    assert cfg.scheduler_sleep_time == 5.0
    assert cfg.interrupt_queue_time == 3600.0
    y = tmp_path / "conf.yaml"
    y.write_text("scheduler_sleep_time: 1.5\nphone_pool:\n  u1:\n    high: 4\n")
    cfg = SimulatorConfig.load(str(y))
    assert cfg.scheduler_sleep_time == 1.5
    assert cfg.phone_pool == {"u1": {"high": 4}}
    assert cfg.timers()["release_sleep_time"] == 10.0


def test_operator_code_staging(tmp_path):
    from olearning_sim_amd.task.staging import (stage_operator_code,
                                                OperatorStagingError)
    # builtin: no staging
    assert stage_operator_code("builtin:fedavg", "train.py", "train",
                               str(tmp_path / "w")) is None
    # directory copy + entry validation
    src = tmp_path / "opsrc"
    src.mkdir()
    (src / "train.py").write_text("print('hi')")
    dst = stage_operator_code(str(src), "train.py", "train",
                              str(tmp_path / "w1"))
    assert os.path.exists(os.path.join(dst, "train.py"))
    with pytest.raises(OperatorStagingError):
        stage_operator_code(str(src), "missing.py", "train",
                            str(tmp_path / "w2"))
    # zip with a wrapping top-level dir gets flattened
    import zipfile
    z = tmp_path / "op.zip"
    with zipfile.ZipFile(z, "w") as zf:
        zf.writestr("mypkg/train.py", "print('hi')")
    dst = stage_operator_code(str(z), "train.py", "train",
                              str(tmp_path / "w3"))
    assert os.path.exists(os.path.join(dst, "train.py"))
    # file-repo key
    from olearning_sim_amd.utils.file_repo import LocalFileRepo
    repo = LocalFileRepo(root=str(tmp_path / "store"))
    repo.upload_file(str(z), "operators", "remote/op.zip")
    dst = stage_operator_code("remote/op.zip", "train.py", "train",
                              str(tmp_path / "w4"), repo=repo)
    assert os.path.exists(os.path.join(dst, "train.py"))


def test_operator_staging_rejects_path_traversal(tmp_path):
    """Operator names from task JSON must never address paths outside
    the task work dir (the resolved target is rmtree'd)."""
    from olearning_sim_amd.task.staging import (stage_operator_code,
                                                OperatorStagingError)
    src = tmp_path / "opsrc"
    src.mkdir()
    (src / "train.py").write_text("print('hi')")
    victim = tmp_path / "victim"
    victim.mkdir()
    (victim / "keep.txt").write_text("data")
    work = tmp_path / "w"
    for bad in ("../victim", "/abs/path", "..", ".", "a/b", "", "a\\b"):
        with pytest.raises(OperatorStagingError):
            stage_operator_code(str(src), "train.py", bad, str(work))
    assert (victim / "keep.txt").exists()
    # zip members that escape the staging dir are rejected
    import zipfile
    z = tmp_path / "evil.zip"
    with zipfile.ZipFile(z, "w") as zf:
        zf.writestr("../escape.py", "print('hi')")
        zf.writestr("train.py", "print('hi')")
    with pytest.raises(OperatorStagingError):
        stage_operator_code(str(z), "train.py", "train", str(tmp_path / "w5"))
    assert not (tmp_path / "escape.py").exists()


def test_validate_rejects_unsafe_names():
    """Submit-time validation rejects operator names / update styles /
    task ids carrying path separators or '..'."""
    import copy
    import json as _json
    from olearning_sim_amd.task import json2taskconfig
    from olearning_sim_amd.task.validate import ValidateParameters
    from test_schema import EXAMPLE

    def check(raw):
        v = ValidateParameters()
        cfg = json2taskconfig(_json.dumps(raw))
        return v.validate_task_parameters(raw, cfg), v.last_error

    assert check(EXAMPLE)[0]
    bad = copy.deepcopy(EXAMPLE)
    bad["task_id"] = "../t1"
    ok, err = check(bad)
    assert not ok and "task_id" in err
    bad = copy.deepcopy(EXAMPLE)
    bad["operatorflow"]["operators"][0]["name"] = "../train"
    ok, err = check(bad)
    assert not ok
    bad = copy.deepcopy(EXAMPLE)
    bad["operatorflow"]["operators"][0]["model"]["model_update_style"] = \
        "../{task_id}_{current_round}.safetensors"
    ok, err = check(bad)
    assert not ok and "model_update_style" in err
    # engine-level guard on the rendered checkpoint name
    from olearning_sim_amd.engine.checkpoint import checkpoint_name
    with pytest.raises(ValueError):
        checkpoint_name("t1", 0, "../{task_id}_{current_round}.bin")


def test_custom_model_plugin(tmp_path, monkeypatch):
    import sys
    (tmp_path / "user_models.py").write_text(
        "from olearning_sim_amd.models.mlp import MLP\n"
        "class TinyNet(MLP):\n"
        "    def __init__(self):\n"
        "        super().__init__(in_features=8, hidden=4, num_classes=2)\n")
    monkeypatch.syspath_prepend(str(tmp_path))
    from olearning_sim_amd.models import build_model
    m = build_model("user_models:TinyNet")
    assert m.num_classes == 2
    gp = m.init_global()
    assert gp["fc1.w"].shape == (8, 4)


def test_release_waits_for_deviceflow_drain():
    import time as _time
    from olearning_sim_amd.deviceflow.service import DeviceFlowService
    from test_manager import make_manager, task_json, wait_terminal
    from olearning_sim_amd.resource.manager import ResourceManager
    from olearning_sim_amd.task.manager import TaskManager
    from olearning_sim_amd.task.runner import TaskRunner
    from olearning_sim_amd.task.table import TaskTableRepo
    svc = DeviceFlowService(time_scale=0.0, seed=0)
    table = TaskTableRepo(":memory:")
    res = ResourceManager(":memory:", totals={"cpu": 8, "mem": 64,
                                              "gpu": 0, "hbm_gb": 0})
    runner = TaskRunner(table, deviceflow=svc)
    mgr = TaskManager(table=table, resource_mgr=res, runner=runner,
                      deviceflow=svc)
    tj = task_json(task_id="t_drain", use_gradient_house=True)
    ok, msg = mgr.submit_task(tj)
    assert ok, msg
    assert mgr.step_schedule() == "t_drain"
    st = wait_terminal(mgr, "t_drain")
    from olearning_sim_amd.task.status import TaskStatus
    assert st == TaskStatus.SUCCEEDED
    t0 = _time.time()
    while _time.time() - t0 < 15:
        if "t_drain" in mgr.step_release():
            break
        _time.sleep(0.05)
    assert not mgr.resources.holding("t_drain")
    assert not svc.registry.is_registered("t_drain")
    svc.shutdown()


def test_engine_resumes_from_latest_checkpoint(tmp_path):
    from olearning_sim_amd.engine import EngineJob, LogicalEngine

    def job(rounds):
        return EngineJob(task_id="rz", model_name="mlp",
                         model_kwargs={"in_features": 16, "hidden": 8,
                                       "num_classes": 4},
                         clients=4, rounds=rounds, local_steps=1,
                         batch_size=2, lr=0.1, num_classes=4, seed=0,
                         checkpoint_dir=str(tmp_path),
                         save_every_round=True)

    e1 = LogicalEngine(job(2))
    e1.run()
    after2 = e1.master.flat.clone()
    # a fresh engine does NOT resume by default (fresh submissions of a
    # task id start at round 0, reference semantics) ...
    assert LogicalEngine(job(4)).start_round == 0
    # ... an explicit crash-recovery restart does
    j = job(4)
    j.resume = True
    e2 = LogicalEngine(j)
    assert e2.start_round == 2
    torch.testing.assert_close(e2.master.flat, after2)
    out = e2.run()
    assert out["rounds"] == 2                 # only rounds 2 and 3 ran
    assert os.path.exists(tmp_path / "rz_3_result_model.safetensors")


def test_drop_all_leaves_master_unchanged():
    import json as _json
    from olearning_sim_amd.engine import EngineJob, LogicalEngine
    strategy = _json.dumps({"real_time_dispatch": {
        "use_strategy": True,
        "drop_simulation": {"drop_probability": 1.0}}})
    job = EngineJob(task_id="dz", model_name="mlp",
                    model_kwargs={"in_features": 16, "hidden": 8,
                                  "num_classes": 4},
                    clients=4, rounds=1, local_steps=1, batch_size=2,
                    lr=0.5, num_classes=4, seed=0,
                    behavior_strategy=strategy, dynamic_num=100)
    eng = LogicalEngine(job)
    before = eng.master.flat.clone()
    rec = eng.run_round(0)
    assert rec["success"] == 4            # clients trained (drops are
    torch.testing.assert_close(eng.master.flat, before)  # post-train)


def _run_cli(args, repo, timeout=150, attempts=2):
    """One retry: CLI subprocesses occasionally hit environment
    transients (slow interpreter start under load); a genuine failure
    fails both attempts."""
    import subprocess, sys
    for attempt in range(attempts):
        out = subprocess.run([sys.executable, "-m", "olearning_sim_amd"]
                             + args, cwd=repo, capture_output=True,
                             text=True, timeout=timeout)
        if out.returncode == 0:
            return out
    return out


@pytest.mark.timeout(400)
def test_cli_submit_wait_and_status(tmp_path):
    import subprocess, sys
    from test_manager import task_json
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    tf = tmp_path / "task.json"
    tf.write_text(task_json(task_id="t_cli"))
    out = _run_cli(["--data-dir", str(tmp_path / "data"),
                    "submit", str(tf), "--wait"], repo)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [json.loads(l) for l in out.stdout.strip().splitlines()]
    assert lines[0]["is_success"]
    assert lines[1]["task_status"] == "SUCCEEDED"
    # status over the SAME data dir sees the terminal state
    out2 = subprocess.run(
        [sys.executable, "-m", "olearning_sim_amd",
         "--data-dir", str(tmp_path / "data"), "status", "t_cli"],
        cwd=repo, capture_output=True, text=True, timeout=60)
    assert json.loads(out2.stdout.strip().splitlines()[-1])[
        "task_status"] == "SUCCEEDED"


@pytest.mark.timeout(400)
def test_cli_result_and_perf(tmp_path):
    from test_manager import task_json
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    tf = tmp_path / "task.json"
    tf.write_text(task_json(task_id="t_cli_r"))
    data_dir = str(tmp_path / "data")
    out = _run_cli(["--data-dir", data_dir, "submit", str(tf), "--wait"],
                   repo)
    assert out.returncode == 0, out.stderr[-2000:]
    res = _run_cli(["--data-dir", data_dir, "result", "t_cli_r"], repo,
                   timeout=90)
    assert res.returncode == 0, res.stderr[-2000:]
    d = json.loads(res.stdout.strip().splitlines()[-1])
    assert d["task_status"] == "SUCCEEDED"
    assert d["logical_result"]["logical_result"][0]["simulation_target"][
        "success_num"]
    perf = _run_cli(["--data-dir", data_dir, "perf", "t_cli_r"], repo,
                    timeout=90)
    assert perf.returncode == 0, perf.stderr[-2000:]
    json.loads(perf.stdout.strip().splitlines()[-1])


def test_sqlite_repo_crud_cycle(tmp_path):
    """Direct CRUD exercise of the generic table repo (the reference's
    SqlDataBase inline self-test, repo_utils.py:346-401)."""
    from olearning_sim_amd.utils.sqlite_repo import SqlTableRepo
    repo = SqlTableRepo(str(tmp_path / "t.sqlite"), "things",
                        {"key": "TEXT", "num": "INTEGER", "blob": "TEXT"},
                        primary_key="key")
    assert repo.add_item({"key": "a", "num": 1, "blob": "x"})
    assert not repo.add_item({"key": "a", "num": 2})   # duplicate pk
    assert repo.has_item("key", "a")
    assert repo.get_item_value("key", "a", "num") == 1
    assert repo.set_item_value("key", "a", "num", 5)
    assert repo.get_item_value("key", "a", "num") == 5
    repo.upsert_item("key", {"key": "b", "num": 7})
    assert repo.count() == 2
    assert sorted(repo.get_column_not_none("key")) == ["a", "b"]
    assert repo.get_rows_where({"num": 7})[0]["key"] == "b"
    assert repo.delete_item("key", "a")
    assert not repo.has_item("key", "a")
    repo.clear()
    assert repo.count() == 0
    repo.close()


def test_sqlite_repo_migrates_new_columns(tmp_path):
    """Reopening a database with a schema that gained columns must
    ALTER the table, not fail on the first write to a new column
    (regression: pre-existing data dirs broke when the task table
    gained the per-side params columns)."""
    from olearning_sim_amd.utils.sqlite_repo import SqlTableRepo
    p = str(tmp_path / "m.sqlite")
    old = SqlTableRepo(p, "t", {"k": "TEXT", "a": "INTEGER"},
                       primary_key="k")
    old.add_item({"k": "x", "a": 1})
    old.close()
    new = SqlTableRepo(p, "t", {"k": "TEXT", "a": "INTEGER",
                                "b": "TEXT", "c": "REAL"},
                       primary_key="k")
    assert new.set_item_value("k", "x", "b", "hello")
    assert new.get_item_value("k", "x", "b") == "hello"
    assert new.get_item_value("k", "x", "a") == 1
    assert new.add_item({"k": "y", "a": 2, "b": "z", "c": 1.5})
    new.close()


@pytest.mark.timeout(300)
def test_examples_run_hermetically(tmp_path):
    """Both examples run green against a fresh data dir (OLSIM_CONFIG
    points the session at tmp) — examples double as integration
    regressions; a stale-schema data dir broke demo_task once."""
    import subprocess, sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cfg = tmp_path / "cfg.yaml"
    cfg.write_text(f"data_dir: {tmp_path / 'data'}\ndevice: cpu\n")
    env = dict(os.environ, OLSIM_CONFIG=str(cfg))
    for script in ("examples/demo_task.py",
                   "examples/submit_script_operator.py"):
        out = subprocess.run([sys.executable, script], cwd=repo, env=env,
                             capture_output=True, text=True, timeout=240)
        assert out.returncode == 0, (script, out.stderr[-2000:])
        assert "SUCCEEDED" in out.stdout, (script, out.stdout[-2000:])


def test_perf_summary_percentiles():
    from olearning_sim_amd.perf.manager import PerformanceManager
    pm = PerformanceManager(":memory:")
    for i, v in enumerate([10.0, 20.0, 30.0, 40.0, 100.0]):
        pm.record("tp", "round_time_s", v, round_idx=i)
    m = pm.summary("tp")["metrics"]["round_time_s"]
    assert m["count"] == 5 and m["last"] == 100.0
    assert m["p50"] == 30.0
    assert m["p95"] == 100.0
    assert m["min"] == 10.0 and m["max"] == 100.0
