"""Submit a task whose per-round work is a user script-file operator
(examples/user_operator/train.py), end to end in-process.

Run: python examples/submit_script_operator.py
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from olearning_sim_amd.session import SimulatorSession

OP_DIR = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "user_operator")

TASK = {
    "user_id": "demo", "task_id": "script_demo",
    "target": {"priority": 5, "data": [{
        "name": "data_0", "data_path": "", "data_split_type": False,
        "data_transfer_type": "FILE", "task_type": "classification",
        "total_simulation": {"devices": ["high"], "nums": [12],
                             "dynamic_nums": [2]},
        "allocation": {"optimization": False,
                       "logical_simulation": [12],
                       "device_simulation": [0],
                       "running_response": {"devices": [], "nums": []}}}]},
    "operatorflow": {
        "flow_setting": {
            "round": 3,
            "start": {"logical_simulation": {"strategy": "",
                                             "wait_interval": 0,
                                             "total_timeout": 0},
                      "device_simulation": {"strategy": "",
                                            "wait_interval": 0,
                                            "total_timeout": 0}},
            "stop": {"logical_simulation": {"strategy": "",
                                            "wait_interval": 0,
                                            "total_timeout": 0},
                     "device_simulation": {"strategy": "",
                                           "wait_interval": 0,
                                           "total_timeout": 0}}},
        "operators": [{
            "name": "train",
            "operation_behavior_controller": {
                "use_gradient_house": False,
                "strategy_gradient_house": "", "outbound_service": ""},
            "input": [], "use_data": True,
            "model": {"use_model": False, "model_for_train": False,
                      "model_transfer_type": "FILE", "model_path": "",
                      "model_update_style": ""},
            "logical_simulation": {
                "operator_transfer_type": "FILE",
                "operator_code_path": OP_DIR,
                "operator_entry_file": "train.py",
                "operator_params": json.dumps({"fail_per_shard": 0})},
            "device_simulation": {"operator_transfer_type": "FILE",
                                  "operator_code_path": "",
                                  "operator_entry_file": "",
                                  "operator_params": ""}}]},
    "logical_simulation": {
        "computation_unit": {"devices": ["high"],
                             "setting": [{"num_cpus": 1}]},
        "resource_request": [{"name": "data_0", "devices": ["high"],
                              "num_request": [2]}]},
    "device_simulation": {"resource_request": []},
}


def main() -> int:
    s = SimulatorSession(svc=0, data_dir="/tmp/ols_script_demo",
                         device="cpu", auto_start_threads=False)
    ok, msg = s.task_mgr.submit_task(json.dumps(TASK))
    print("submit:", ok, msg)
    assert ok, msg
    assert s.task_mgr.step_schedule() == "script_demo"
    while not s.task_mgr.get_task_status("script_demo").is_terminal():
        time.sleep(0.2)
    s.task_mgr.step_release()
    print("final status:", s.task_mgr.get_task_status("script_demo").value)
    row = s.task_mgr.table.get_row("script_demo")
    print("logical_result:", row["logical_result"])
    s.shutdown()
    return 0


if __name__ == "__main__":
    sys.exit(main())
