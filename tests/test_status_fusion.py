"""Table-driven tests of the status-fusion truth table
(reference task_manager.py:610-889 semantics)."""

import json

import pytest

from olearning_sim_amd.resource.manager import ResourceManager
from olearning_sim_amd.task.manager import TaskManager
from olearning_sim_amd.task.runner import TaskRunner
from olearning_sim_amd.task.status import TaskStatus
from olearning_sim_amd.task.table import TaskTableRepo


def mk():
    table = TaskTableRepo(":memory:")
    mgr = TaskManager(table=table,
                      resource_mgr=ResourceManager(
                          ":memory:", totals={"cpu": 8, "mem": 8, "gpu": 0,
                                              "hbm_gb": 0}),
                      runner=TaskRunner(table))
    return mgr, table


def seed(table, task_id="t", max_round=2, nums=(10,), dynamic=(2,),
         logical=True, device=False):
    table.add_task(task_id)
    total = {"max_round": max_round, "operator_name_list": ["train"],
             "data_name_list": ["d0"],
             "total_simulation": [{"name": "d0", "simulation_target": {
                 "devices": ["high"], "nums": list(nums),
                 "dynamic_nums": list(dynamic)}}]}
    table.set_items(task_id, total_simulation=json.dumps(total))
    if logical:
        table.set_item_value(task_id, "logical_target", json.dumps(
            {"logical_target": [{"name": "d0", "simulation_target": {
                "devices": ["high"], "nums": list(nums)}}]}))
    if device:
        table.set_item_value(task_id, "device_target", json.dumps(
            {"device_target": [{"name": "d0", "simulation_target": {
                "devices": ["high"], "nums": list(nums)}}]}))


def set_logical_result(table, task_id, success, failed, rnd, op="train"):
    table.set_items(task_id,
                    logical_result=json.dumps({"logical_result": [
                        {"name": "d0", "simulation_target": {
                            "devices": ["high"], "success_num": [success],
                            "failed_num": [failed]}}]}),
                    logical_round=rnd, logical_operator=op)


def test_success_when_tolerance_met():
    mgr, table = mk()
    seed(table)
    set_logical_result(table, "t", success=8, failed=2, rnd=2)
    st = mgr.combine_task_status("t", TaskStatus.SUCCEEDED,
                                 {"is_finished": True, "device_result": []})
    assert st == TaskStatus.SUCCEEDED


def test_success_exactly_at_threshold():
    mgr, table = mk()
    seed(table, nums=(10,), dynamic=(2,))
    set_logical_result(table, "t", success=8, failed=0, rnd=2)
    assert mgr.combine_task_status(
        "t", TaskStatus.SUCCEEDED,
        {"is_finished": True, "device_result": []}) == TaskStatus.SUCCEEDED


def test_failure_below_threshold_after_last_round():
    mgr, table = mk()
    seed(table)
    set_logical_result(table, "t", success=7, failed=3, rnd=2)
    st = mgr.combine_task_status("t", TaskStatus.SUCCEEDED,
                                 {"is_finished": True, "device_result": []})
    assert st == TaskStatus.FAILED


def test_round_failure_fails_early():
    mgr, table = mk()
    seed(table)   # dynamic=2
    set_logical_result(table, "t", success=5, failed=3, rnd=1)  # 3 > 2
    st = mgr.combine_task_status("t", TaskStatus.RUNNING,
                                 {"is_finished": True, "device_result": []})
    assert st == TaskStatus.FAILED


def test_running_mid_rounds():
    mgr, table = mk()
    seed(table)
    set_logical_result(table, "t", success=9, failed=1, rnd=1)
    st = mgr.combine_task_status("t", TaskStatus.RUNNING,
                                 {"is_finished": False, "device_result": []})
    assert st == TaskStatus.RUNNING


def test_no_result_yet_is_running():
    mgr, table = mk()
    seed(table)
    st = mgr.combine_task_status("t", TaskStatus.RUNNING,
                                 {"is_finished": False, "device_result": []})
    assert st == TaskStatus.RUNNING


def test_stopped_job_with_no_success_is_stopped():
    mgr, table = mk()
    seed(table)
    set_logical_result(table, "t", success=4, failed=1, rnd=1)
    st = mgr.combine_task_status("t", TaskStatus.STOPPED,
                                 {"is_finished": True, "device_result": []})
    assert st == TaskStatus.STOPPED


def test_logical_job_failed_fails_task():
    mgr, table = mk()
    seed(table)
    set_logical_result(table, "t", success=4, failed=0, rnd=1)
    st = mgr.combine_task_status("t", TaskStatus.FAILED,
                                 {"is_finished": True, "device_result": []})
    assert st == TaskStatus.FAILED


def test_hybrid_success_sums_both_sides():
    """logical 6 + device 4 successes cover nums=10 with dynamic=2."""
    mgr, table = mk()
    seed(table, logical=True, device=True)
    set_logical_result(table, "t", success=6, failed=0, rnd=2)
    table.set_items("t", device_round=2, device_operator="train")
    device_result = {"is_finished": True, "device_result": [
        {"name": "d0", "simulation_target": {"devices": ["high"],
                                             "success_num": [4],
                                             "failed_num": [0]}}]}
    st = mgr.combine_task_status("t", TaskStatus.SUCCEEDED, device_result)
    assert st == TaskStatus.SUCCEEDED


def test_hybrid_combined_failures_break_tolerance():
    mgr, table = mk()
    seed(table, logical=True, device=True)   # dynamic=2
    set_logical_result(table, "t", success=4, failed=2, rnd=1)
    table.set_items("t", device_round=1, device_operator="train")
    device_result = {"is_finished": False, "device_result": [
        {"name": "d0", "simulation_target": {"devices": ["high"],
                                             "success_num": [2],
                                             "failed_num": [1]}}]}
    # combined failed = 3 > dynamic 2 at the same round -> early failure
    st = mgr.combine_task_status("t", TaskStatus.RUNNING, device_result)
    assert st == TaskStatus.FAILED


def test_device_only_task():
    mgr, table = mk()
    seed(table, logical=False, device=True)
    table.set_items("t", device_round=2, device_operator="train")
    device_result = {"is_finished": True, "device_result": [
        {"name": "d0", "simulation_target": {"devices": ["high"],
                                             "success_num": [9],
                                             "failed_num": [1]}}]}
    st = mgr.combine_task_status("t", None, device_result)
    assert st == TaskStatus.SUCCEEDED


def test_missing_total_simulation_fails():
    mgr, table = mk()
    table.add_task("t")
    st = mgr.combine_task_status("t", TaskStatus.RUNNING,
                                 {"is_finished": True, "device_result": []})
    assert st == TaskStatus.FAILED


def test_missing_data_entry_is_not_success():
    """A targeted data with no result entry must block SUCCEEDED (the
    lenient skip would silently pass multi-data tasks that only
    reported one data)."""
    mgr, table = mk()
    table.add_task("t2")
    total = {"max_round": 1, "operator_name_list": ["train"],
             "data_name_list": ["d0", "d1"],
             "total_simulation": [
                 {"name": "d0", "simulation_target": {
                     "devices": ["high"], "nums": [10],
                     "dynamic_nums": [2]}},
                 {"name": "d1", "simulation_target": {
                     "devices": ["high"], "nums": [5],
                     "dynamic_nums": [0]}}]}
    table.set_items("t2", total_simulation=json.dumps(total))
    table.set_item_value("t2", "logical_target", json.dumps(
        {"logical_target": [
            {"name": "d0", "simulation_target": {"devices": ["high"],
                                                 "nums": [10]}},
            {"name": "d1", "simulation_target": {"devices": ["high"],
                                                 "nums": [5]}}]}))
    # only d0 reported
    table.set_items("t2", logical_result=json.dumps({"logical_result": [
        {"name": "d0", "simulation_target": {
            "devices": ["high"], "success_num": [10],
            "failed_num": [0]}}]}),
        logical_round=1, logical_operator="train")
    st = mgr.combine_task_status("t2", TaskStatus.SUCCEEDED,
                                 {"is_finished": True})
    assert st != TaskStatus.SUCCEEDED


def test_short_success_vector_is_not_success():
    """A result vector missing a targeted tier must not pass via zip
    truncation."""
    mgr, table = mk()
    seed(table, "t3", max_round=1, nums=(10, 5), dynamic=(2, 0))
    # only one tier reported for a two-tier target
    table.set_items("t3", logical_result=json.dumps({"logical_result": [
        {"name": "d0", "simulation_target": {
            "devices": ["high"], "success_num": [10],
            "failed_num": [0]}}]}),
        logical_round=1, logical_operator="train")
    st = mgr.combine_task_status("t3", TaskStatus.SUCCEEDED,
                                 {"is_finished": True})
    assert st != TaskStatus.SUCCEEDED
