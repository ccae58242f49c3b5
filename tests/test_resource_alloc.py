"""Resource ledger + hybrid allocation + scheduler-strategy tests."""

import json

from olearning_sim_amd.resource.manager import ResourceManager
from olearning_sim_amd.task.allocation import (HybridOptimizer, ALPHA, BETA,
                                               LAMBDA, _cost_logical,
                                               _cost_device)
from olearning_sim_amd.task.schema import json2taskconfig
from olearning_sim_amd.task.scheduler import DefaultStrategy


def test_resource_request_release_cycle():
    rm = ResourceManager(":memory:", totals={"cpu": 8, "mem": 32, "gpu": 2,
                                             "hbm_gb": 576},
                         phone_pool={"u1": {"high": 5, "low": 10}})
    assert rm.request_resource("t1", "u1", cpu=4, mem=16, gpu=1, hbm_gb=288,
                               phones={"high": 3})
    remain = rm.get_remain_res()
    assert remain["cpu"] == 4 and remain["gpu"] == 1
    avail = rm.get_resource("u1")["device_simulation"]["u1"]
    assert avail == {"high": 2, "low": 10}
    # over-request denied
    assert not rm.request_resource("t2", "u1", cpu=5)
    assert not rm.request_resource("t3", "u1", phones={"high": 3})
    # double-hold denied
    assert not rm.request_resource("t1", "u1", cpu=1)
    rm.release_resource("t1")
    assert rm.get_remain_res()["cpu"] == 8
    assert rm.request_resource("t2", "u1", cpu=5)


def _cfg(optimization, nums=1000, actors=4, phones=10, rr=0):
    return json2taskconfig(json.dumps({
        "user_id": "u", "task_id": "t",
        "target": {"data": [{
            "name": "d0",
            "total_simulation": {"devices": ["high"], "nums": [nums],
                                 "dynamic_nums": [0]},
            "allocation": {"optimization": optimization,
                           "logical_simulation": [nums - 100],
                           "device_simulation": [100],
                           "running_response": {"devices": ["high"],
                                                "nums": [rr]}}}]},
        "operatorflow": {"flow_setting": {"round": 1}, "operators": []},
        "logical_simulation": {
            "computation_unit": {"devices": ["high"],
                                 "setting": [{"num_cpus": 1}]},
            "resource_request": [{"name": "d0", "devices": ["high"],
                                  "num_request": [actors]}]},
        "device_simulation": {
            "resource_request": [{"name": "d0", "devices": ["high"],
                                  "num_request": [phones]}]},
    }))


def test_fixed_allocation_respected():
    allocs = HybridOptimizer(_cfg(False)).allocate()
    t = allocs[0].tiers[0]
    assert (t.logical, t.device) == (900, 100)


def test_auto_allocation_balances_makespan():
    allocs = HybridOptimizer(_cfg(True, nums=1000, actors=4, phones=10)).allocate()
    t = allocs[0].tiers[0]
    assert t.logical + t.device == 1000
    assert t.logical > 0 and t.device > 0
    # optimum should beat both all-logical and all-device
    best = max(_cost_logical(t.logical, 4), _cost_device(t.device, 10))
    assert best <= max(_cost_logical(1000, 4), _cost_device(0, 10)) + 1e-9
    assert best <= max(_cost_logical(0, 4), _cost_device(1000, 10)) + 1e-9
    # device side is much faster per unit (BETA << ALPHA), so it should
    # carry the bulk of the machine-times
    assert t.device > t.logical


def test_auto_allocation_no_phones_goes_logical():
    allocs = HybridOptimizer(_cfg(True, phones=0)).allocate()
    t = allocs[0].tiers[0]
    assert (t.logical, t.device) == (1000, 0)


def test_auto_allocation_respects_running_response():
    allocs = HybridOptimizer(_cfg(True, rr=50)).allocate()
    t = allocs[0].tiers[0]
    assert t.device >= 50


def test_strategy_scoring_prefers_priority_then_position():
    waiting = [{"task_priority": 0}, {"task_priority": 0},
               {"task_priority": 10}]
    assert DefaultStrategy.schedule_task(waiting) == 2
    waiting = [{"task_priority": 3}, {"task_priority": 3}]
    assert DefaultStrategy.schedule_task(waiting) == 0  # earlier wins ties


def test_request_resource_formula():
    cfg = _cfg(False)
    req = DefaultStrategy.get_task_request_resource(cfg)
    assert req["logical_simulation"]["cpu"] == 4.0   # 4 units x 1 cpu
    assert req["device_simulation"]["u"] == {"high": 10}
