"""Per-side task assembly + device-task conversion
(reference utils_runner.py:478-628, 784-932 parity)."""

import copy
import json

from olearning_sim_amd.task import json2taskconfig
from olearning_sim_amd.task.allocation import HybridOptimizer
from olearning_sim_amd.task.submitter import (JobSubmitter, fix_device_task_json,
                                              json2deviceconfig)

from test_schema import EXAMPLE


def hybrid_task():
    raw = copy.deepcopy(EXAMPLE)
    raw["target"]["data"][0]["allocation"] = {
        "optimization": False,
        "logical_simulation": [70, 40],
        "device_simulation": [30, 10],
        "running_response": {"devices": [], "nums": []}}
    raw["device_simulation"] = {"resource_request": [
        {"name": "data_0", "devices": ["high", "low"],
         "num_request": [10, 5]}]}
    return json2taskconfig(json.dumps(raw))


def make_submitter():
    task = hybrid_task()
    return JobSubmitter(task, HybridOptimizer(task).allocate())


def test_logical_side_gets_its_split_and_paths():
    side = make_submitter().assemble_info_logical_simulation()
    d = side["target"]["data"][0]
    assert d["data_path"] == "bucket/data_0_logical.zip"
    assert d["total_simulation"]["nums"] == [70, 40]
    # tolerance scales with population: floor(5*70/100), floor(2*40/50)
    assert d["total_simulation"]["dynamic_nums"] == [3, 1]
    assert d["allocation"]["logical_simulation"] == [70, 40]
    assert d["allocation"]["device_simulation"] == [0, 0]
    # the other side's operator info is stripped
    op = side["operatorflow"]["operators"][0]
    assert op["device_simulation"]["operator_code_path"] == ""
    assert op["logical_simulation"]["operator_code_path"] == "builtin:fedavg"


def test_device_side_gets_its_split():
    side = make_submitter().assemble_info_device_simulation()
    d = side["target"]["data"][0]
    assert d["data_path"] == "bucket/data_0_device.zip"
    assert d["total_simulation"]["nums"] == [30, 10]
    assert d["total_simulation"]["dynamic_nums"] == [1, 0]
    assert side["operatorflow"]["operators"][0][
        "logical_simulation"]["operator_entry_file"] == ""


def test_sides_partition_the_population():
    sub = make_submitter()
    log = sub.assemble_info_logical_simulation()["target"]["data"][0]
    dev = sub.assemble_info_device_simulation()["target"]["data"][0]
    orig = EXAMPLE["target"]["data"][0]["total_simulation"]
    for i in range(2):
        assert (log["total_simulation"]["nums"][i]
                + dev["total_simulation"]["nums"][i]) == orig["nums"][i]
        assert (log["total_simulation"]["dynamic_nums"][i]
                + dev["total_simulation"]["dynamic_nums"][i]) \
            <= orig["dynamic_nums"][i]


def test_pure_logical_task_has_no_device_side():
    task = json2taskconfig(json.dumps(EXAMPLE))   # all-logical allocation
    sub = JobSubmitter(task, HybridOptimizer(task).allocate())
    assert sub.assemble_info_device_simulation() is None
    assert sub.assemble_info_logical_simulation() is not None


def test_json2deviceconfig_and_fix():
    side = make_submitter().assemble_info_device_simulation()
    cfg = json2deviceconfig(side)
    assert cfg["task_id"] == "task_001"
    assert cfg["rounds"] == 4
    assert cfg["data"][0]["nums"] == [30, 10]
    assert cfg["operators"][0]["name"] == "train"

    fixed = fix_device_task_json(cfg, "task_001", "ws://gh.example/out")
    p = json.loads(fixed["operators"][0]["params"])
    assert p["task_id"] == "task_001"
    # outbound only rewritten for gradient-house operators
    assert fixed["operators"][0].get("outbound_service", "") == ""


def test_runner_persists_per_side_params():
    """TaskManager lifecycle leaves the assembled per-side configs in
    the task table (reference writes logical/device task JSON to DB)."""
    import sys
    sys.path.insert(0, "tests")
    from test_manager import make_manager, task_json, wait_terminal
    mgr = make_manager()
    ok, msg = mgr.submit_task(task_json(task_id="t_side"))
    assert ok, msg
    assert mgr.step_schedule() == "t_side"
    st = wait_terminal(mgr, "t_side")
    assert st.value == "SUCCEEDED"
    raw = mgr.table.get_item_value("t_side", "logical_task_params")
    assert raw
    side = json.loads(raw)
    d = side["target"]["data"][0]
    assert sum(d["total_simulation"]["nums"]) == 6
    mgr.shutdown()
