#!/usr/bin/env python3
"""End-to-end demo: submit a federated-simulation task through the full
control plane and watch it run.

    python examples/demo_task.py             # CPU, small MLP task
    python examples/demo_task.py --gpu       # bf16 LeNet on cuda:0
"""

import argparse
import json
import sys
import os
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from olearning_sim_amd.session import SimulatorSession


def make_task(gpu: bool) -> dict:
    if gpu:
        op_params = {"model": "lenet", "model_kwargs": {"num_classes": 10},
                     "lr": 0.05, "local_steps": 2, "batch_size": 16,
                     "num_classes": 10, "dtype": "bfloat16"}
        clients = 200
    else:
        op_params = {"model": "mlp",
                     "model_kwargs": {"in_features": 64, "hidden": 32,
                                      "num_classes": 10},
                     "lr": 0.1, "local_steps": 2, "batch_size": 8,
                     "num_classes": 10}
        clients = 20
    return {
        "user_id": "demo", "task_id": "demo_task",
        "target": {"priority": 5, "data": [{
            "name": "data_0", "data_path": "", "data_split_type": False,
            "data_transfer_type": "FILE", "task_type": "classification",
            "total_simulation": {"devices": ["high"], "nums": [clients],
                                 "dynamic_nums": [2]},
            "allocation": {"optimization": False,
                           "logical_simulation": [clients],
                           "device_simulation": [0],
                           "running_response": {"devices": [], "nums": []}}}]},
        "operatorflow": {
            "flow_setting": {"round": 3,
                             "start": {"logical_simulation": {},
                                       "device_simulation": {}},
                             "stop": {"logical_simulation": {},
                                      "device_simulation": {}}},
            "operators": [{
                "name": "train",
                "operation_behavior_controller": {
                    "use_gradient_house": True,
                    "strategy_gradient_house": json.dumps(
                        {"real_time_dispatch": {"use_strategy": True,
                                                "dispatch_batch_sizes": [8]}}),
                    "outbound_service": ""},
                "input": [], "use_data": True,
                "model": {"use_model": True, "model_for_train": True,
                          "model_path": "demo", "model_update_style":
                          "{task_id}_{current_round}_result_model.safetensors"},
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
                                  "num_request": [4]}]},
        "device_simulation": {"resource_request": []},
    }


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpu", action="store_true")
    args = ap.parse_args()

    sess = SimulatorSession(svc=0, auto_start_threads=True,
                            device="cuda:0" if args.gpu else "cpu")
    ok, msg = sess.task_mgr.submit_task(json.dumps(make_task(args.gpu)))
    print(f"submit: {ok} ({msg})")
    while True:
        status = sess.task_mgr.get_task_status("demo_task")
        print(f"status: {status.value}  queue={sess.task_mgr.get_task_queue()}")
        if status.is_terminal():
            break
        time.sleep(1.0)
    print("perf summary:",
          json.dumps(sess.performance_mgr.summary("demo_task"), indent=2))
    row = sess.task_mgr.table.get_row("demo_task")
    print("logical_result:", row["logical_result"])
    print("checkpoints under:",
          os.path.join(sess.data_dir, "checkpoints"))
    sess.shutdown()
    return 0


if __name__ == "__main__":
    sys.exit(main())
