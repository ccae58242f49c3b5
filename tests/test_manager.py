"""End-to-end task lifecycle through the control plane:
submit -> validate -> queue -> schedule -> freeze -> engine run ->
status fusion -> release. (reference call stack SURVEY.md §3.1/§3.4)"""

import json
import time

import pytest

from olearning_sim_amd.resource.manager import ResourceManager
from olearning_sim_amd.task.manager import TaskManager
from olearning_sim_amd.task.runner import TaskRunner
from olearning_sim_amd.task.status import TaskStatus
from olearning_sim_amd.task.table import TaskTableRepo


def task_json(task_id="t_e2e", rounds=2, clients=6, dynamic=1,
              priority=0, params=None, use_gradient_house=False):
    op_params = {
        "model": "mlp",
        "model_kwargs": {"in_features": 32, "hidden": 16, "num_classes": 5},
        "lr": 0.1, "local_steps": 1, "batch_size": 4, "num_classes": 5,
        "shard_size": 8,
    }
    if params:
        op_params.update(params)
    return json.dumps({
        "user_id": "u1", "task_id": task_id,
        "target": {"priority": priority, "data": [{
            "name": "data_0", "data_path": "", "data_split_type": False,
            "data_transfer_type": "FILE", "task_type": "classification",
            "total_simulation": {"devices": ["high"], "nums": [clients],
                                 "dynamic_nums": [dynamic]},
            "allocation": {"optimization": False,
                           "logical_simulation": [clients],
                           "device_simulation": [0],
                           "running_response": {"devices": [], "nums": []}}}]},
        "operatorflow": {
            "flow_setting": {"round": rounds,
                             "start": {"logical_simulation": {}, "device_simulation": {}},
                             "stop": {"logical_simulation": {}, "device_simulation": {}}},
            "operators": [{
                "name": "train",
                "operation_behavior_controller": {
                    "use_gradient_house": use_gradient_house,
                    "strategy_gradient_house": json.dumps(
                        {"real_time_dispatch": {"use_strategy": True}})
                    if use_gradient_house else "",
                    "outbound_service": ""},
                "input": [], "use_data": True,
                "model": {"use_model": False},
                "logical_simulation": {
                    "operator_transfer_type": "FILE",
                    "operator_code_path": "builtin:fedavg",
                    "operator_entry_file": "train.py",
                    "operator_params": json.dumps(op_params)},
                "device_simulation": {}}]},
        "logical_simulation": {
            "computation_unit": {"devices": ["high"],
                                 "setting": [{"num_cpus": 1}]},
            "resource_request": [{"name": "data_0", "devices": ["high"],
                                  "num_request": [2]}]},
        "device_simulation": {"resource_request": []},
    })


def make_manager(cpu=8.0):
    table = TaskTableRepo(":memory:")
    res = ResourceManager(":memory:", totals={"cpu": cpu, "mem": 64.0,
                                              "gpu": 0, "hbm_gb": 0})
    runner = TaskRunner(table, device="cpu")
    return TaskManager(table=table, resource_mgr=res, runner=runner)


def wait_terminal(mgr, task_id, timeout=30.0):
    t0 = time.time()
    while time.time() - t0 < timeout:
        st = mgr.get_task_status(task_id)
        if st.is_terminal():
            return st
        time.sleep(0.05)
    return mgr.get_task_status(task_id)


def test_submit_validates_and_queues():
    mgr = make_manager()
    ok, msg = mgr.submit_task(task_json())
    assert ok, msg
    assert mgr.get_task_status("t_e2e") == TaskStatus.QUEUED
    assert mgr.get_task_queue() == ["t_e2e"]
    # duplicate rejected
    ok2, msg2 = mgr.submit_task(task_json())
    assert not ok2


def test_submit_rejects_invalid():
    mgr = make_manager()
    bad = json.loads(task_json())
    bad["target"]["priority"] = 99
    ok, msg = mgr.submit_task(json.dumps(bad))
    assert not ok and "priority" in msg


def test_full_lifecycle_succeeds():
    mgr = make_manager()
    ok, _ = mgr.submit_task(task_json())
    assert ok
    task_id = mgr.step_schedule()
    assert task_id == "t_e2e"
    assert mgr.table.get_item_value("t_e2e", "resource_occupied") == 1
    assert mgr.resources.holding("t_e2e")
    st = wait_terminal(mgr, "t_e2e")
    assert st == TaskStatus.SUCCEEDED
    # release step frees the quota
    released = mgr.step_release()
    assert "t_e2e" in released
    assert not mgr.resources.holding("t_e2e")
    # results recorded with reference shapes
    lr = json.loads(mgr.table.get_item_value("t_e2e", "logical_result"))
    tgt = lr["logical_result"][0]["simulation_target"]
    assert tgt["success_num"] == [6]
    assert mgr.table.get_item_value("t_e2e", "logical_round") == 2


def test_no_schedule_when_resources_missing():
    mgr = make_manager(cpu=1.0)   # needs 2 cpus
    mgr.submit_task(task_json())
    assert mgr.step_schedule() is None
    assert mgr.get_task_status("t_e2e") == TaskStatus.QUEUED


def test_priority_scheduling_order():
    mgr = make_manager()
    mgr.submit_task(task_json(task_id="low", priority=0))
    mgr.submit_task(task_json(task_id="high", priority=10))
    first = mgr.step_schedule()
    assert first == "high"


def test_stop_queued_task():
    mgr = make_manager(cpu=1.0)
    mgr.submit_task(task_json())
    ok, _ = mgr.stop_task("t_e2e")
    assert ok
    assert mgr.get_task_status("t_e2e") == TaskStatus.STOPPED


def test_missing_task_status():
    mgr = make_manager()
    assert mgr.get_task_status("nope") == TaskStatus.MISSING


def test_interrupt_overdue_queued_task():
    mgr = make_manager(cpu=1.0)
    mgr.timers["interrupt_queue_time"] = 0.0
    mgr.submit_task(task_json())
    time.sleep(0.02)
    assert "t_e2e" in mgr.step_interrupt()
    assert mgr.get_task_status("t_e2e") == TaskStatus.FAILED


def test_round_failure_reports_failed():
    """All clients offline + tolerance 0 -> round fails -> task FAILED."""
    mgr = make_manager()
    strategy = json.dumps({"offline_simulation": {"offline_probability": 1.0}})
    tj = json.loads(task_json(task_id="t_fail", dynamic=1))
    op = tj["operatorflow"]["operators"][0]
    op["operation_behavior_controller"]["use_gradient_house"] = True
    op["operation_behavior_controller"]["strategy_gradient_house"] = strategy
    mgr.submit_task(json.dumps(tj))
    assert mgr.step_schedule() == "t_fail"
    st = wait_terminal(mgr, "t_fail")
    assert st == TaskStatus.FAILED


def test_queue_recovery_from_table():
    table = TaskTableRepo(":memory:")
    res = ResourceManager(":memory:", totals={"cpu": 8, "mem": 64,
                                              "gpu": 0, "hbm_gb": 0})
    mgr1 = TaskManager(table=table, resource_mgr=res,
                       runner=TaskRunner(table))
    mgr1.submit_task(task_json())
    # simulate restart: new manager over the same table
    mgr2 = TaskManager(table=table, resource_mgr=res,
                       runner=TaskRunner(table))
    assert mgr2.get_task_queue() == ["t_e2e"]


def test_multi_operator_task_lifecycle():
    """train -> evaluate operator list: fusion requires the LAST
    operator (reference calculate_conditions last-operator check)."""
    mgr = make_manager()
    tj = json.loads(task_json(task_id="t_multi", rounds=2))
    ops = tj["operatorflow"]["operators"]
    ops.append({
        "name": "evaluate",
        "operation_behavior_controller": {"use_gradient_house": False,
                                          "strategy_gradient_house": "",
                                          "outbound_service": ""},
        "input": ["train"], "use_data": False,
        "model": {"use_model": False},
        "logical_simulation": {
            "operator_transfer_type": "FILE",
            "operator_code_path": "builtin:evaluate",
            "operator_entry_file": "eval.py",
            "operator_params": json.dumps({"kind": "evaluate"})},
        "device_simulation": {}})
    ok, msg = mgr.submit_task(json.dumps(tj))
    assert ok, msg
    assert mgr.step_schedule() == "t_multi"
    st = wait_terminal(mgr, "t_multi")
    assert st == TaskStatus.SUCCEEDED
    assert mgr.table.get_item_value("t_multi", "logical_operator") == "evaluate"


def test_stop_running_task_reports_stopped():
    mgr = make_manager()
    # long-ish task: 60 rounds of a small model
    tj = json.loads(task_json(task_id="t_stop", rounds=60, clients=6))
    mgr.submit_task(json.dumps(tj))
    assert mgr.step_schedule() == "t_stop"
    # let it get going, then stop
    t0 = time.time()
    while time.time() - t0 < 10:
        if (mgr.table.get_item_value("t_stop", "logical_round") or 0) >= 1:
            break
        time.sleep(0.02)
    ok, _ = mgr.stop_task("t_stop")
    assert ok
    st = wait_terminal(mgr, "t_stop", timeout=30)
    assert st == TaskStatus.STOPPED
    assert "t_stop" in mgr.step_release()
    assert not mgr.resources.holding("t_stop")


def hybrid_task_json(task_id, logical=4, device=2, phones=3):
    tj = json.loads(task_json(task_id=task_id, clients=logical + device,
                              dynamic=1))
    d = tj["target"]["data"][0]
    d["allocation"]["logical_simulation"] = [logical]
    d["allocation"]["device_simulation"] = [device]
    tj["device_simulation"]["resource_request"] = [
        {"name": "data_0", "devices": ["high"], "num_request": [phones]}]
    return json.dumps(tj)


def test_hybrid_task_with_simulated_phone_farm():
    """Hybrid allocation: logical engine + simulated phone farm both run
    and the fused status combines their per-tier successes."""
    from olearning_sim_amd.resource.manager import ResourceManager
    table = TaskTableRepo(":memory:")
    res = ResourceManager(":memory:", totals={"cpu": 8, "mem": 64,
                                              "gpu": 0, "hbm_gb": 0},
                          phone_pool={"u1": {"high": 8}})
    mgr = TaskManager(table=table, resource_mgr=res,
                      runner=TaskRunner(table))
    ok, msg = mgr.submit_task(hybrid_task_json("t_hybrid"))
    assert ok, msg
    assert mgr.step_schedule() == "t_hybrid"
    st = wait_terminal(mgr, "t_hybrid", timeout=60)
    assert st == TaskStatus.SUCCEEDED
    dr = json.loads(table.get_item_value("t_hybrid", "device_result"))
    assert dr["device_result"][0]["simulation_target"]["success_num"] == [2]
    lr = json.loads(table.get_item_value("t_hybrid", "logical_result"))
    assert lr["logical_result"][0]["simulation_target"]["success_num"] == [4]


def test_three_tasks_contend_for_resources_and_all_finish():
    """Only 2 tasks fit at once (4 cpus / 2 each): the scheduler runs
    them as resources free up; all three succeed."""
    mgr = make_manager(cpu=4.0)
    for tid in ("a1", "a2", "a3"):
        ok, msg = mgr.submit_task(task_json(task_id=tid, rounds=1))
        assert ok, msg
    done = set()
    t0 = time.time()
    while len(done) < 3 and time.time() - t0 < 60:
        mgr.step_schedule()
        for tid in ("a1", "a2", "a3"):
            st = mgr.get_task_status(tid)
            if st.is_terminal():
                done.add(tid)
        mgr.step_release()
        time.sleep(0.05)
    assert done == {"a1", "a2", "a3"}
    for tid in done:
        assert mgr.get_task_status(tid) == TaskStatus.SUCCEEDED
        assert not mgr.resources.holding(tid)


def test_multi_data_task_reports_each_data():
    """A task with two data targets gets per-data success/failed
    vectors in logical_result (reference analyze_results accumulates
    per data name, run_task.py:149-210)."""
    import copy
    raw = json.loads(task_json(task_id="t_multi", rounds=2, clients=6,
                               dynamic=1))
    d1 = copy.deepcopy(raw["target"]["data"][0])
    d1["name"] = "data_1"
    d1["total_simulation"] = {"devices": ["high"], "nums": [4],
                              "dynamic_nums": [1]}
    d1["allocation"] = {"optimization": False, "logical_simulation": [4],
                        "device_simulation": [0],
                        "running_response": {"devices": [], "nums": []}}
    raw["target"]["data"].append(d1)
    raw["logical_simulation"]["resource_request"].append(
        {"name": "data_1", "devices": ["high"], "num_request": [1]})
    mgr = make_manager()
    ok, msg = mgr.submit_task(json.dumps(raw))
    assert ok, msg
    assert mgr.step_schedule() == "t_multi"
    st = wait_terminal(mgr, "t_multi")
    assert st == TaskStatus.SUCCEEDED
    res = json.loads(mgr.table.get_row("t_multi")["logical_result"])
    entries = {e["name"]: e["simulation_target"]
               for e in res["logical_result"]}
    assert entries["data_0"]["success_num"] == [6]
    assert entries["data_1"]["success_num"] == [4]
    mgr.shutdown()


def test_interrupt_overdue_running_task():
    """The interrupt watchdog also stops tasks running past
    interrupt_running_time (reference task_manager.py:1150-1200)."""
    mgr = make_manager()
    mgr.timers["interrupt_running_time"] = 0.0
    # enough rounds that the engine cannot finish before the watchdog
    # fires even on a loaded host (the stop lands within a round or two)
    mgr.submit_task(task_json(task_id="t_long", rounds=5000))
    assert mgr.step_schedule() == "t_long"
    time.sleep(0.05)
    assert "t_long" in mgr.step_interrupt()
    st = wait_terminal(mgr, "t_long")
    assert st in (TaskStatus.STOPPED, TaskStatus.FAILED)
    mgr.shutdown()


def test_engine_crash_marks_task_failed():
    """A crash inside the engine (reference: Ray job failure) fails the
    task instead of wedging it RUNNING."""
    raw = json.loads(task_json(task_id="t_crash"))
    op = raw["operatorflow"]["operators"][0]
    params = json.loads(op["logical_simulation"]["operator_params"])
    params["model"] = "no_such_model"
    op["logical_simulation"]["operator_params"] = json.dumps(params)
    mgr = make_manager()
    ok, msg = mgr.submit_task(json.dumps(raw))
    assert ok, msg
    assert mgr.step_schedule() == "t_crash"
    st = wait_terminal(mgr, "t_crash")
    assert st == TaskStatus.FAILED
    # release step frees the frozen resources
    mgr.step_release()
    assert mgr.table.get_item_value("t_crash", "resource_occupied") == 0
    mgr.shutdown()


def test_resubmit_with_resume_continues_from_checkpoint(tmp_path):
    """operator_params resume:true picks up the newest templated
    checkpoint of the same task id (reference: actors download round
    r-1 weights; task state persists across manager restarts)."""
    from olearning_sim_amd.resource.manager import ResourceManager
    from olearning_sim_amd.task.runner import TaskRunner
    from olearning_sim_amd.task.table import TaskTableRepo
    from olearning_sim_amd.task.manager import TaskManager
    ckpt = str(tmp_path / "ck")
    table = TaskTableRepo(":memory:")
    res = ResourceManager(":memory:", totals={"cpu": 8.0, "mem": 64.0,
                                              "gpu": 0, "hbm_gb": 0})
    mgr = TaskManager(table=table, resource_mgr=res,
                      runner=TaskRunner(table, device="cpu",
                                        checkpoint_dir=ckpt))
    raw = json.loads(task_json(task_id="t_res2", rounds=2))
    op = raw["operatorflow"]["operators"][0]
    op["model"] = {"use_model": True, "model_for_train": True,
                   "model_transfer_type": "FILE", "model_path": "m",
                   "model_update_style":
                       "{task_id}_{current_round}_result_model.safetensors"}
    ok, msg = mgr.submit_task(json.dumps(raw))
    assert ok, msg
    assert mgr.step_schedule() == "t_res2"
    assert wait_terminal(mgr, "t_res2") == TaskStatus.SUCCEEDED
    mgr.step_release()
    import os
    assert os.path.exists(os.path.join(
        ckpt, "t_res2_1_result_model.safetensors"))

    # resubmit for 4 rounds with resume: engine continues at round 2
    round1 = os.path.join(ckpt, "t_res2_1_result_model.safetensors")
    mtime_before = os.path.getmtime(round1)
    raw["operatorflow"]["flow_setting"]["round"] = 4
    p = json.loads(op["logical_simulation"]["operator_params"])
    p["resume"] = True
    op["logical_simulation"]["operator_params"] = json.dumps(p)
    ok, msg = mgr.submit_task(json.dumps(raw))
    assert ok, msg
    assert mgr.step_schedule() == "t_res2"
    assert wait_terminal(mgr, "t_res2") == TaskStatus.SUCCEEDED
    # final cursor reaches max_round and rounds 2..3 were saved
    assert table.get_item_value("t_res2", "logical_round") == 4
    assert os.path.exists(os.path.join(
        ckpt, "t_res2_3_result_model.safetensors"))
    # proof of resume: round 1's artifact was NOT rewritten
    assert os.path.getmtime(round1) == mtime_before
    mgr.shutdown()


@pytest.mark.timeout(120)
def test_threaded_manager_runs_concurrent_tasks():
    """With the real daemon loops on (fast timers), several tasks
    schedule, run, and release concurrently without manual stepping —
    the reference's 3-thread operation."""
    table = TaskTableRepo(":memory:")
    res = ResourceManager(":memory:", totals={"cpu": 4.0, "mem": 64.0,
                                              "gpu": 0, "hbm_gb": 0})
    mgr = TaskManager(table=table, resource_mgr=res,
                      runner=TaskRunner(table, device="cpu"),
                      timers={"scheduler_sleep_time": 0.05,
                              "release_sleep_time": 0.05,
                              "interrupt_sleep_time": 60.0},
                      auto_start=True)
    try:
        for tid in ("th1", "th2", "th3"):
            ok, msg = mgr.submit_task(task_json(task_id=tid, rounds=1))
            assert ok, msg
        t0 = time.time()
        while time.time() - t0 < 60:
            done = {tid for tid in ("th1", "th2", "th3")
                    if mgr.get_task_status(tid) == TaskStatus.SUCCEEDED
                    and table.get_item_value(tid, "resource_occupied") == 0}
            if len(done) == 3:
                break
            time.sleep(0.05)
        assert len(done) == 3, {
            tid: mgr.get_task_status(tid).value for tid in
            ("th1", "th2", "th3")}
    finally:
        mgr.shutdown()


def test_orphaned_running_task_fails_on_restart(tmp_path):
    """A task RUNNING when the process died must not poll as RUNNING
    forever in the next process — it fails and its resources free."""
    db = str(tmp_path / "t.sqlite")
    rdb = str(tmp_path / "r.sqlite")
    table = TaskTableRepo(db)
    res = ResourceManager(rdb, totals={"cpu": 8, "mem": 64, "gpu": 0,
                                       "hbm_gb": 0})
    mgr1 = TaskManager(table=table, resource_mgr=res,
                       runner=TaskRunner(table))
    ok, _ = mgr1.submit_task(task_json(task_id="t_orph", rounds=1))
    assert ok
    # simulate crash mid-run: row says RUNNING + resources frozen, but
    # the next process's runner has no such job
    res.request_resource("t_orph", "u1", cpu=2.0)
    table.set_items("t_orph", task_status=TaskStatus.RUNNING.value,
                    resource_occupied=1, job_id="olsjob_dead")
    mgr1.shutdown()

    table2 = TaskTableRepo(db)
    res2 = ResourceManager(rdb, totals={"cpu": 8, "mem": 64, "gpu": 0,
                                        "hbm_gb": 0})
    mgr2 = TaskManager(table=table2, resource_mgr=res2,
                       runner=TaskRunner(table2))
    assert mgr2.get_task_status("t_orph") == TaskStatus.FAILED
    assert not res2.holding("t_orph")
    # and the task can be resubmitted fresh
    ok, msg = mgr2.submit_task(task_json(task_id="t_orph", rounds=1))
    assert ok, msg
    mgr2.shutdown()


def test_submit_routes_multi_gpu_task_to_worker_group(tmp_path):
    """A task whose train operator asks num_gpus > 1 runs as a worker
    group (one process per device, torchrun; gloo on CPU) launched via
    the NodeClusterManager, and its result rows fuse to SUCCEEDED
    (reference TaskRunner submits to the Ray fabric,
    task_runner.py:41-87)."""
    from olearning_sim_amd.cluster.node_manager import NodeClusterManager
    table = TaskTableRepo(":memory:")
    res = ResourceManager(":memory:", totals={"cpu": 8.0, "mem": 64.0,
                                              "gpu": 0, "hbm_gb": 0})
    cluster = NodeClusterManager()
    runner = TaskRunner(table, device="cpu",
                        checkpoint_dir=str(tmp_path), cluster=cluster)
    mgr = TaskManager(table=table, resource_mgr=res, runner=runner)
    try:
        ok, msg = mgr.submit_task(task_json(
            task_id="t_grp", rounds=2, clients=6,
            params={"num_gpus": 2}))
        assert ok, msg
        assert mgr.step_schedule() == "t_grp"
        st = wait_terminal(mgr, "t_grp", timeout=180.0)
        assert st == TaskStatus.SUCCEEDED, \
            f"status={st} err={[ (h.job_id, h.error) for h in runner.jobs.values() ]}"
        jids = runner.task_jobs["t_grp"]
        assert any(j.startswith("olsgrp_") for j in jids)
        # rank 0's round rows were replayed into the table
        assert int(table.get_item_value("t_grp", "logical_round")) == 2
        result = json.loads(table.get_item_value("t_grp", "logical_result"))
        entry = result["logical_result"][0]["simulation_target"]
        assert sum(entry["success_num"]) == 6      # both ranks' shards
        # the group was reaped
        assert cluster.list_clusters() == []
    finally:
        mgr.shutdown()
        cluster.shutdown()
