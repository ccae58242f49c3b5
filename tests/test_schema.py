"""Task JSON schema round-trip (reference utils.py:831-1197 parity)."""

import json

from olearning_sim_amd.task import json2taskconfig, taskconfig2json

EXAMPLE = {
    "user_id": "user1",
    "task_id": "task_001",
    "target": {
        "priority": 3,
        "data": [{
            "name": "data_0",
            "data_path": "bucket/data_0.zip",
            "data_split_type": True,
            "data_transfer_type": "MINIO",
            "task_type": "classification",
            "total_simulation": {"devices": ["high", "low"],
                                 "nums": [100, 50], "dynamic_nums": [5, 2]},
            "allocation": {"optimization": False,
                           "logical_simulation": [100, 50],
                           "device_simulation": [0, 0],
                           "running_response": {"devices": [], "nums": []}},
        }]},
    "operatorflow": {
        "flow_setting": {
            "round": 4,
            "start": {"logical_simulation": {"strategy": "", "wait_interval": 0,
                                             "total_timeout": 0},
                      "device_simulation": {"strategy": "", "wait_interval": 0,
                                            "total_timeout": 0}},
            "stop": {"logical_simulation": {"strategy": "", "wait_interval": 0,
                                            "total_timeout": 0},
                     "device_simulation": {"strategy": "", "wait_interval": 0,
                                           "total_timeout": 0}}},
        "operators": [{
            "name": "train",
            "operation_behavior_controller": {"use_gradient_house": False,
                                              "strategy_gradient_house": "",
                                              "outbound_service": ""},
            "input": [],
            "use_data": True,
            "model": {"use_model": True, "model_for_train": True,
                      "model_transfer_type": "MINIO",
                      "model_path": "bucket/model",
                      "model_update_style": "{task_id}_{current_round}_result_model.safetensors"},
            "logical_simulation": {"operator_transfer_type": "FILE",
                                   "operator_code_path": "builtin:fedavg",
                                   "operator_entry_file": "train.py",
                                   "operator_params": "{\"lr\": 0.05}"},
            "device_simulation": {"operator_transfer_type": "S3",
                                  "operator_code_path": "",
                                  "operator_entry_file": "",
                                  "operator_params": ""}}]},
    "logical_simulation": {
        "computation_unit": {"devices": ["high", "low"],
                             "setting": [{"num_cpus": 1}, {"num_cpus": 1}]},
        "resource_request": [{"name": "data_0", "devices": ["high", "low"],
                              "num_request": [4, 2]}]},
    "device_simulation": {"resource_request": []},
}


def test_roundtrip_preserves_all_fields():
    cfg = json2taskconfig(json.dumps(EXAMPLE))
    back = json.loads(taskconfig2json(cfg))
    assert back == json.loads(json.dumps(EXAMPLE))


def test_defaults_applied():
    cfg = json2taskconfig(json.dumps({"task_id": "t", "user_id": "u",
                                      "target": {"data": [{}]}}))
    d = cfg.target.data[0]
    assert d.name == "data_0"           # default name = data_<index>
    assert d.data_transfer_type == "S3"
    assert cfg.target.priority == 0
    assert cfg.operatorflow.flow_setting.round == 0


def test_parsed_values():
    cfg = json2taskconfig(json.dumps(EXAMPLE))
    assert cfg.task_id == "task_001"
    assert cfg.target.data[0].total_simulation.nums == [100, 50]
    assert cfg.operatorflow.operators[0].model.use_model is True
    assert cfg.logical_simulation.computation_unit.setting[0].num_cpus == 1
