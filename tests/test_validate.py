"""Three-stage task validation (reference utils.py:283-811 parity)."""

import copy
import json

import pytest

from olearning_sim_amd.task import json2taskconfig
from olearning_sim_amd.task.validate import ValidateParameters

from test_schema import EXAMPLE


def check(raw):
    v = ValidateParameters()
    cfg = json2taskconfig(json.dumps(raw))
    ok = v.validate_task_parameters(raw, cfg)
    return ok, v.last_error


def test_example_valid():
    ok, err = check(EXAMPLE)
    assert ok, err


def mutate(path, value):
    raw = copy.deepcopy(EXAMPLE)
    node = raw
    for key in path[:-1]:
        node = node[key]
    node[path[-1]] = value
    return raw


@pytest.mark.parametrize("path,value", [
    (("user_id",), ""),                                  # empty user
    (("task_id",), "任务"),                               # non-ASCII
    (("target", "priority"), 11),                        # out of range
    (("target", "data", 0, "total_simulation", "nums"), [0, 50]),   # nums<=0
    (("target", "data", 0, "total_simulation", "dynamic_nums"), [-1, 2]),
    (("target", "data", 0, "total_simulation", "devices"), ["high", "high"]),
    (("target", "data", 0, "data_transfer_type"), "FTP"),
    (("operatorflow", "flow_setting", "round"), 0),
    (("operatorflow", "operators", 0, "name"), "has space"),
    (("operatorflow", "operators", 0, "logical_simulation", "operator_params"),
     "not json"),
    (("operatorflow", "operators", 0, "logical_simulation", "operator_entry_file"),
     "train.txt"),
    (("logical_simulation", "computation_unit", "setting"),
     [{"num_cpus": 0}, {"num_cpus": 1}]),
])
def test_correctness_rejections(path, value):
    ok, err = check(mutate(path, value))
    assert not ok
    assert err


@pytest.mark.parametrize("path,value", [
    # nums must exceed dynamic_nums
    (("target", "data", 0, "total_simulation", "dynamic_nums"), [100, 50]),
    # devices/nums/dynamic_nums same length
    (("target", "data", 0, "total_simulation", "nums"), [100]),
    # allocation must sum to nums when optimization False
    (("target", "data", 0, "allocation", "logical_simulation"), [90, 50]),
    # running_response devices must be subset of devices
    (("target", "data", 0, "allocation", "running_response"),
     {"devices": ["phantom"], "nums": [1]}),
    # wait_interval <= total_timeout
    (("operatorflow", "flow_setting", "start", "logical_simulation"),
     {"strategy": "", "wait_interval": 5, "total_timeout": 1}),
    # input must reference earlier operators
    (("operatorflow", "operators", 0, "input"), ["missing_op"]),
    # model_path required when use_model
    (("operatorflow", "operators", 0, "model"),
     {"use_model": True, "model_path": ""}),
    # resource_request names must equal target data names
    (("logical_simulation", "resource_request"),
     [{"name": "other", "devices": ["high"], "num_request": [1]}]),
    # computation_unit must cover all tiers
    (("logical_simulation", "computation_unit"),
     {"devices": ["high"], "setting": [{"num_cpus": 1}]}),
    # gradient house needs a strategy
    (("operatorflow", "operators", 0, "operation_behavior_controller"),
     {"use_gradient_house": True, "strategy_gradient_house": ""}),
    # code paths not both empty
    (("operatorflow", "operators", 0, "logical_simulation", "operator_code_path"), ""),
])
def test_relationship_rejections(path, value):
    ok, err = check(mutate(path, value))
    assert not ok
    assert err


def test_type_rejections():
    ok, err = check(mutate(("target", "data", 0, "total_simulation", "nums"),
                           ["a", "b"]))
    assert not ok
    ok, err = check(mutate(("operatorflow", "operators", 0, "use_data"), "yes"))
    assert not ok


def test_non_minio_data_path_must_be_zip():
    raw = mutate(("target", "data", 0, "data_transfer_type"), "S3")
    raw["target"]["data"][0]["data_path"] = "bucket/plaindir"
    ok, err = check(raw)
    assert not ok and "zip" in err.lower()
    raw["target"]["data"][0]["data_path"] = "bucket/data.zip"
    ok, err = check(raw)
    assert ok, err


def test_dir_code_path_requires_file_transfer(tmp_path):
    d = tmp_path / "opdir"
    d.mkdir()
    raw = mutate(("operatorflow", "operators", 0, "logical_simulation"),
                 {"operator_transfer_type": "S3",
                  "operator_code_path": str(d),
                  "operator_entry_file": "train.py",
                  "operator_params": "{}"})
    ok, err = check(raw)
    assert not ok and "FILE" in err
