"""Script-file operator execution (reference Actor subprocess path,
utils_run_task.py:481-514, staged via utils_runner.py:684-782)."""

import json
import os
import sys
import textwrap
import time

import pytest

sys.path.insert(0, os.path.dirname(__file__))

OP_SCRIPT = textwrap.dedent("""
    import json, os, sys

    p = json.loads(sys.argv[sys.argv.index("--params") + 1])
    n = p["actor_simulation_num"]
    lo, hi = p["client_range"]
    # deterministic failure: the shard holding client 0 fails one device
    # in round 0
    fail = 1 if lo == 0 and p["current_round"] == 0 else 0
    extra = json.loads(p["operator"]["operator_params"] or "{}")
    if extra.get("fail_all"):
        fail = n
    with open(os.path.join(p["actor_save_dir"], "result.json"), "w") as f:
        json.dump({"success": n - fail, "failed": fail}, f)
""")


@pytest.fixture
def op_dir(tmp_path):
    d = tmp_path / "user_op"
    d.mkdir()
    (d / "train.py").write_text(OP_SCRIPT)
    return d


def test_script_operator_counts_and_ranges(op_dir, tmp_path):
    from olearning_sim_amd.engine.script_op import ScriptOperator, per_tier_counts
    op = ScriptOperator(name="user_train", staged_dir=str(op_dir),
                        entry_file="train.py", operator_params="{}",
                        task_id="t_s", work_dir=str(tmp_path / "w"),
                        clients=10, shards=2)
    assert op.shard_ranges() == [(0, 5), (5, 10)]
    res = op.run_round(0)
    assert res["success"] == 9 and res["failed"] == 1
    assert res["failed_ranges"] == [(4, 5)]      # tail of shard 0
    succ_t, fail_t = per_tier_counts(res["failed_ranges"], [0, 6, 10])
    assert succ_t == [5, 4] and fail_t == [1, 0]
    res1 = op.run_round(1)
    assert res1["failed"] == 0


def test_script_operator_nonzero_exit_fails_shard(tmp_path):
    d = tmp_path / "bad_op"
    d.mkdir()
    (d / "train.py").write_text("import sys; sys.exit(3)\n")
    from olearning_sim_amd.engine.script_op import ScriptOperator
    op = ScriptOperator(name="bad", staged_dir=str(d), entry_file="train.py",
                        operator_params="", task_id="t", work_dir=str(tmp_path / "w"),
                        clients=4, shards=2)
    res = op.run_round(0)
    assert res["success"] == 0 and res["failed"] == 4


def test_engine_runs_script_operator(op_dir, tmp_path):
    from olearning_sim_amd.engine import EngineJob, LogicalEngine
    from olearning_sim_amd.engine.script_op import ScriptOperator
    job = EngineJob(task_id="t_se", model_name="mlp",
                    model_kwargs={"in_features": 16, "hidden": 8,
                                  "num_classes": 4},
                    clients=8, rounds=2, local_steps=1, batch_size=4,
                    lr=0.1, device="cpu", dtype="float32", num_classes=4,
                    dynamic_num=2,
                    operators=[("user_train", "script")])
    sop = ScriptOperator(name="user_train", staged_dir=str(op_dir),
                         entry_file="train.py", operator_params="{}",
                         task_id="t_se", work_dir=str(tmp_path / "w"),
                         clients=8, shards=2)
    eng = LogicalEngine(job, script_ops={"user_train": sop})
    out = eng.run()
    assert out["rounds"] == 2
    assert out["success_total"] == 8 + 7   # round 1 clean, round 0 one fail
    assert out["failed_total"] == 1


def test_task_lifecycle_with_script_operator(op_dir):
    """Full submit→schedule→script-run→fuse path through TaskManager."""
    from test_manager import make_manager, task_json, wait_terminal
    raw = json.loads(task_json(task_id="t_script", rounds=2, clients=8,
                               dynamic=2))
    op = raw["operatorflow"]["operators"][0]
    op["logical_simulation"]["operator_transfer_type"] = "FILE"
    op["logical_simulation"]["operator_code_path"] = str(op_dir)
    op["logical_simulation"]["operator_entry_file"] = "train.py"
    op["logical_simulation"]["operator_params"] = "{}"
    mgr = make_manager()
    ok, msg = mgr.submit_task(json.dumps(raw))
    assert ok, msg
    assert mgr.step_schedule() == "t_script"
    st = wait_terminal(mgr, "t_script")
    assert st.value == "SUCCEEDED"
    row = mgr.table.get_row("t_script")
    res = json.loads(row["logical_result"])
    tgt = res["logical_result"][0]["simulation_target"]
    assert sum(tgt["success_num"]) == 8    # final round: all succeed
    mgr.shutdown()


def test_script_operator_failure_fails_round(op_dir):
    from test_manager import make_manager, task_json, wait_terminal
    raw = json.loads(task_json(task_id="t_sfail", rounds=3, clients=8,
                               dynamic=1))
    op = raw["operatorflow"]["operators"][0]
    op["logical_simulation"]["operator_transfer_type"] = "FILE"
    op["logical_simulation"]["operator_code_path"] = str(op_dir)
    op["logical_simulation"]["operator_entry_file"] = "train.py"
    op["logical_simulation"]["operator_params"] = json.dumps(
        {"fail_all": True})
    mgr = make_manager()
    ok, msg = mgr.submit_task(json.dumps(raw))
    assert ok, msg
    assert mgr.step_schedule() == "t_sfail"
    st = wait_terminal(mgr, "t_sfail")
    assert st.value == "FAILED"
    mgr.shutdown()


def test_script_operator_drives_deviceflow(op_dir, tmp_path):
    """Gradient-house lifecycle wraps script operators too: NotifyStart,
    a round summary through the message plane, NotifyComplete."""
    import time as _time
    from olearning_sim_amd.deviceflow.service import DeviceFlowService
    from olearning_sim_amd.engine import EngineJob, LogicalEngine
    from olearning_sim_amd.engine.script_op import ScriptOperator
    strategy = json.dumps({"real_time_dispatch": {
        "use_strategy": True, "dispatch_batch_sizes": [1]}})
    svc = DeviceFlowService(time_scale=0.0, seed=1)
    svc.register_task("t_sdf", ["logical_simulation"])
    job = EngineJob(task_id="t_sdf", model_name="mlp",
                    model_kwargs={"in_features": 16, "hidden": 8,
                                  "num_classes": 4},
                    clients=4, rounds=2, local_steps=1, batch_size=2,
                    lr=0.1, device="cpu", dtype="float32", num_classes=4,
                    dynamic_num=4, behavior_strategy=strategy,
                    operators=[("user_train", "script")])
    sop = ScriptOperator(name="user_train", staged_dir=str(op_dir),
                         entry_file="train.py", operator_params="{}",
                         task_id="t_sdf", work_dir=str(tmp_path / "w"),
                         clients=4, shards=1)
    eng = LogicalEngine(job, script_ops={"user_train": sop},
                        deviceflow=svc, behavior=lambda r, c: (
                            __import__("torch").zeros(c, dtype=bool),
                            __import__("torch").zeros(c, dtype=bool)))
    eng.run()
    t0 = _time.time()
    while _time.time() - t0 < 10 and not svc.check_dispatch_finished("t_sdf"):
        _time.sleep(0.01)
    assert svc.check_dispatch_finished("t_sdf")
    msgs = svc.outbound.drain()
    assert len(msgs) == 2                       # one summary per round
    assert {m.payload["round"] for m in msgs} == {0, 1}
    assert all(m.payload["success"] in (3, 4) for m in msgs)
    svc.shutdown()


def test_script_operator_receives_model_path(tmp_path):
    """Round r>0 passes the previous round's templated checkpoint path
    to the script (reference download_model_files semantics)."""
    import textwrap as _tw
    d = tmp_path / "mp_op"
    d.mkdir()
    (d / "train.py").write_text(_tw.dedent("""
        import json, os, sys
        p = json.loads(sys.argv[sys.argv.index("--params") + 1])
        out = {"success": p["actor_simulation_num"], "failed": 0,
               "model_path": p["operator"]["model"].get("current_model_path")}
        with open(os.path.join(p["actor_save_dir"], "result.json"), "w") as f:
            json.dump(out, f)
    """))
    from olearning_sim_amd.engine import EngineJob, LogicalEngine
    from olearning_sim_amd.engine.script_op import ScriptOperator
    ckpt = str(tmp_path / "ckpt")
    job = EngineJob(task_id="t_mp", model_name="mlp",
                    model_kwargs={"in_features": 16, "hidden": 8,
                                  "num_classes": 4},
                    clients=4, rounds=2, local_steps=1, batch_size=2,
                    lr=0.1, device="cpu", dtype="float32", num_classes=4,
                    dynamic_num=4, checkpoint_dir=ckpt,
                    save_every_round=True,
                    operators=[("user_train", "script"),
                               ("save", "checkpoint")])
    wdir = tmp_path / "w"
    sop = ScriptOperator(name="user_train", staged_dir=str(d),
                         entry_file="train.py", operator_params="{}",
                         task_id="t_mp", work_dir=str(wdir),
                         clients=4, shards=1)
    LogicalEngine(job, script_ops={"user_train": sop}).run()
    r0 = json.loads((wdir / "round_0" / "shard_0" / "result.json").read_text())
    r1 = json.loads((wdir / "round_1" / "shard_0" / "result.json").read_text())
    assert r0["model_path"] is None
    assert r1["model_path"] and r1["model_path"].endswith(
        "t_mp_0_result_model.safetensors")
    assert os.path.exists(r1["model_path"])
