"""Capstone integration: every major feature in ONE task, end to end
through the JSON API — two-tier population, gradient house with churn
and real-time dispatch, multi-operator round (train + evaluate +
checkpoint), per-round checkpoints under the templated name, per-tier
result vectors, perf metrics, dispatch curves, and resource release."""

import copy
import json
import os
import sys
import time

import pytest
from fastapi.testclient import TestClient

from olearning_sim_amd.api.server import build_app
from olearning_sim_amd.session import SimulatorSession

sys.path.insert(0, os.path.dirname(__file__))


def capstone_task(task_id="t_cap", rounds=3):
    strategy = json.dumps({
        "real_time_dispatch": {"use_strategy": True,
                               "dispatch_batch_sizes": [2]},
        "offline_simulation": {"offline_probability": 0.25},
    })
    op_params = {
        "model": "mlp",
        "model_kwargs": {"in_features": 32, "hidden": 16, "num_classes": 5},
        "lr": 0.1, "local_steps": 1, "batch_size": 4, "num_classes": 5,
        "shard_size": 8,
    }
    empty_cond = {"strategy": "", "wait_interval": 0, "total_timeout": 0}

    def operator(name, kind=None):
        p = dict(op_params)
        if kind:
            p["kind"] = kind
        return {
            "name": name,
            "operation_behavior_controller": {
                "use_gradient_house": name == "train",
                "strategy_gradient_house": strategy if name == "train" else "",
                "outbound_service": ""},
            "input": [] if name == "train" else ["train"],
            "use_data": True,
            "model": {"use_model": True, "model_for_train": True,
                      "model_transfer_type": "FILE", "model_path": "m",
                      "model_update_style":
                          "{task_id}_{current_round}_result_model.safetensors"},
            "logical_simulation": {
                "operator_transfer_type": "FILE",
                "operator_code_path": f"builtin:{name}",
                "operator_entry_file": "train.py",
                "operator_params": json.dumps(p)},
            "device_simulation": {"operator_transfer_type": "FILE",
                                  "operator_code_path": "",
                                  "operator_entry_file": "",
                                  "operator_params": ""}}

    return {
        "user_id": "cap", "task_id": task_id,
        "target": {"priority": 7, "data": [{
            "name": "data_0", "data_path": "", "data_split_type": False,
            "data_transfer_type": "FILE", "task_type": "classification",
            "total_simulation": {"devices": ["high", "low"],
                                 "nums": [8, 4], "dynamic_nums": [7, 3]},
            "allocation": {"optimization": False,
                           "logical_simulation": [8, 4],
                           "device_simulation": [0, 0],
                           "running_response": {"devices": [], "nums": []}}}]},
        "operatorflow": {
            "flow_setting": {"round": rounds,
                             "start": {"logical_simulation": empty_cond,
                                       "device_simulation": empty_cond},
                             "stop": {"logical_simulation": empty_cond,
                                      "device_simulation": empty_cond}},
            "operators": [operator("train"),
                          operator("evaluate", "evaluate"),
                          operator("save_model", "checkpoint")]},
        "logical_simulation": {
            "computation_unit": {"devices": ["high", "low"],
                                 "setting": [{"num_cpus": 1},
                                             {"num_cpus": 1}]},
            "resource_request": [{"name": "data_0",
                                  "devices": ["high", "low"],
                                  "num_request": [2, 1]}]},
        "device_simulation": {"resource_request": []},
    }


@pytest.mark.timeout(180)
def test_capstone_everything_in_one_task(tmp_path):
    session = SimulatorSession(svc=0, data_dir=str(tmp_path), device="cpu",
                               auto_start_threads=False)
    try:
        client = TestClient(build_app(session))
        task = capstone_task()
        r = client.post("/taskmgr/submitTask", json={"task": task}).json()
        assert r["is_success"], r

        assert session.task_mgr.step_schedule() == "t_cap"
        t0 = time.time()
        while time.time() - t0 < 90:
            st = client.get("/taskmgr/getTaskStatus/t_cap").json()[
                "task_status"]
            if st in ("SUCCEEDED", "FAILED", "STOPPED"):
                break
            time.sleep(0.1)
        assert st == "SUCCEEDED", st

        # per-tier result vectors for both tiers, final round, last operator
        res = client.get("/taskmgr/getTaskResult/t_cap").json()
        assert res["logical_round"] == 3
        tgt = res["logical_result"]["logical_result"][0]["simulation_target"]
        assert tgt["devices"] == ["high", "low"]
        assert tgt["success_num"][0] + tgt["failed_num"][0] == 8
        assert tgt["success_num"][1] + tgt["failed_num"][1] == 4

        # per-round checkpoints under the templated name
        ckpt_dir = os.path.join(str(session.data_dir), "checkpoints")
        ckpts = [f for f in os.listdir(ckpt_dir) if f.startswith("t_cap_")]
        assert ("t_cap_2_result_model.safetensors" in ckpts), ckpts

        # gradient house: flows drained, dispatch curves recorded
        assert client.get(
            "/deviceflow/CheckDeviceflowDispatchFinished/t_cap").json()[
            "is_finished"]
        curves = client.get("/deviceflow/dispatchCurve/t_cap").json()["flows"]
        assert curves, "no dispatch curves recorded"
        sent = sum(row["sent"] for c in curves.values() for row in c)
        assert sent > 0

        # perf metrics recorded per round
        perf = client.get("/performancemgr/summary/t_cap").json()
        assert perf.get("rounds", 0) >= 3 or perf  # summary shape varies

        # release: resources freed, deviceflow unregistered
        released = session.task_mgr.step_release()
        assert "t_cap" in released
        row = session.task_mgr.table.get_row("t_cap")
        assert row["resource_occupied"] == 0
        assert row["finish_task_time"] is not None

        # per-side assembled config persisted at submit
        side = json.loads(row["logical_task_params"])
        assert side["target"]["data"][0]["total_simulation"]["nums"] == [8, 4]
    finally:
        session.shutdown()
