"""Submit a task over the real gRPC wire and poll it to completion.

The server side is `SimulatorSession.serve_grpc()` — the six services
of the reference's proto surface on actual protobuf wire format.  This
client uses the runtime-compiled message classes; a client generated
from the reference's .proto files with stock protoc produces identical
bytes.

    python examples/grpc_client.py
"""

import json
import sys
import os
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import grpc

from olearning_sim_amd.session import SimulatorSession
from olearning_sim_amd.api.grpc_server import (registry,
                                               taskconfig_json_to_msg)

TASK = {
    "user_id": "demo", "task_id": "t_grpc_demo",
    "target": {"priority": 0, "data": [{
        "name": "data_0", "data_path": "", "data_split_type": False,
        "data_transfer_type": "FILE", "task_type": "classification",
        "total_simulation": {"devices": ["high"], "nums": [8],
                             "dynamic_nums": [1]},
        "allocation": {"optimization": False, "logical_simulation": [8],
                       "device_simulation": [0],
                       "running_response": {"devices": [], "nums": []}}}]},
    "operatorflow": {
        "flow_setting": {"round": 2,
                         "start": {"logical_simulation": {},
                                   "device_simulation": {}},
                         "stop": {"logical_simulation": {},
                                  "device_simulation": {}}},
        "operators": [{
            "name": "train",
            "operation_behavior_controller": {"use_gradient_house": False},
            "input": [], "use_data": True, "model": {"use_model": False},
            "logical_simulation": {
                "operator_transfer_type": "FILE",
                "operator_code_path": "builtin:fedavg",
                "operator_entry_file": "train.py",
                "operator_params": json.dumps({
                    "model": "mlp",
                    "model_kwargs": {"in_features": 16, "hidden": 8,
                                     "num_classes": 4},
                    "lr": 0.1, "local_steps": 1, "batch_size": 4,
                    "num_classes": 4, "shard_size": 8})},
            "device_simulation": {}}]},
    "logical_simulation": {
        "computation_unit": {"devices": ["high"],
                             "setting": [{"num_cpus": 1}]},
        "resource_request": [{"name": "data_0", "devices": ["high"],
                              "num_request": [1]}]},
    "device_simulation": {"resource_request": []},
}

STATUS_NAMES = {0: "SUCCEEDED", 1: "PENDING", 2: "RUNNING", 3: "STOPPED",
                4: "FAILED", 5: "MISSING", 6: "UNDONE", 7: "QUEUED"}


def main():
    session = SimulatorSession(svc=0, device="cpu")
    server = session.serve_grpc()
    reg = registry()
    chan = grpc.insecure_channel(f"127.0.0.1:{server._ols_port}")
    OpStatus = reg.msg("taskService.proto", "OperationStatus")
    TaskStatusM = reg.msg("taskService.proto", "TaskStatus")
    TaskID = reg.msg("taskService.proto", "TaskID")

    def call(method, msg, resp_cls):
        fn = chan.unary_unary(
            f"/TaskMgr/{method}",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=resp_cls.FromString)
        return fn(msg, timeout=10)

    out = call("submitTask", taskconfig_json_to_msg(TASK), OpStatus)
    print("submitTask is_success:", out.is_success)

    deadline = time.time() + 60
    while time.time() < deadline:
        st = call("getTaskStatus", TaskID(taskID="t_grpc_demo"), TaskStatusM)
        name = STATUS_NAMES.get(st.taskStatus, "?")
        print("status:", name)
        if name in ("SUCCEEDED", "FAILED", "STOPPED"):
            break
        time.sleep(0.3)
    server.stop(0)
    session.shutdown()
    assert name == "SUCCEEDED", name
    print("done: task ran to completion over the gRPC wire")


if __name__ == "__main__":
    main()
