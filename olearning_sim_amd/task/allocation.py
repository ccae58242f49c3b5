"""Hybrid logical/device machine-time allocation.

Parity with the reference's HybridOptimizer
(ols_core/taskMgr/utils/utils_runner.py:23-176) and its PuLP CBC
integer program `auto_allocation_hybrid_task` (:939-1022): each target
data's machine-times per device tier are split between logical
simulation (server-side virtual devices) and device simulation (real
phones) so the slower side's makespan is minimised, under the cost
model

  logical time ≈ ALPHA * ceil(n_logical / actors)
  device  time ≈ LAMBDA + BETA * n_device / phones

with the reference's published constants ALPHA=3.5 s, BETA=0.14 s,
LAMBDA=8.808 s (utils_runner.py:941-943).  Instead of shipping a MILP
solver, the optimum of this two-resource makespan split is found by
direct search over the (single-dimensional per tier, jointly convex)
split — exact for the same cost model.

When `allocation.optimization` is False the user's fixed split is used
unchanged (reference hybrid_setting_by_user, :53-80).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List

from .schema import TaskConfig

ALPHA = 3.5     # seconds per logical machine-time round (reference CPU fleet)
BETA = 0.14     # seconds per real-device machine-time round
LAMBDA = 8.808  # real-device startup cost, seconds


@dataclass
class TierAllocation:
    tier: str
    total: int
    logical: int
    device: int
    running_response: int = 0


@dataclass
class DataAllocation:
    data_name: str
    tiers: List[TierAllocation] = field(default_factory=list)

    def logical_total(self) -> int:
        return sum(t.logical for t in self.tiers)

    def device_total(self) -> int:
        return sum(t.device for t in self.tiers)


def _cost_logical(n: int, actors: int) -> float:
    if n <= 0:
        return 0.0
    actors = max(1, actors)
    return ALPHA * ((n + actors - 1) // actors)

def _cost_device(n: int, phones: int) -> float:
    if n <= 0:
        return 0.0
    phones = max(1, phones)
    return LAMBDA + BETA * n / phones


class HybridOptimizer:
    """Compute the per-data, per-tier logical/device split."""

    def __init__(self, config: TaskConfig):
        self.config = config

    def _resources(self, data_name: str) -> Dict[str, Dict[str, int]]:
        out = {"logical": {}, "device": {}}
        for rr in self.config.logical_simulation.resource_request:
            if rr.name == data_name:
                out["logical"] = dict(zip(rr.devices, rr.num_request))
        for rr in self.config.device_simulation.resource_request:
            if rr.name == data_name:
                out["device"] = dict(zip(rr.devices, rr.num_request))
        return out

    def allocate(self) -> List[DataAllocation]:
        result = []
        for data in self.config.target.data:
            ts = data.total_simulation
            rr_map = dict(zip(data.allocation.running_response.devices,
                              data.allocation.running_response.nums))
            alloc = DataAllocation(data_name=data.name)
            if not data.allocation.optimization:
                la = list(data.allocation.logical_simulation) or [0] * len(ts.nums)
                da = list(data.allocation.device_simulation) or [0] * len(ts.nums)
                for i, tier in enumerate(ts.devices):
                    alloc.tiers.append(TierAllocation(
                        tier=tier, total=ts.nums[i], logical=la[i],
                        device=da[i], running_response=rr_map.get(tier, 0)))
            else:
                res = self._resources(data.name)
                for i, tier in enumerate(ts.devices):
                    n = ts.nums[i]
                    actors = res["logical"].get(tier, 0)
                    phones = res["device"].get(tier, 0)
                    rr = rr_map.get(tier, 0)
                    alloc.tiers.append(self._optimize_tier(
                        tier, n, actors, phones, rr))
            result.append(alloc)
        return result

    @staticmethod
    def _optimize_tier(tier: str, n: int, actors: int, phones: int,
                       rr: int) -> TierAllocation:
        """Exact minimiser of max(logical, device) cost for one tier.

        Device side must carry at least the running-response
        machine-times (the 2024-10-23 constraint in
        utils_runner.py / utils.py:605-621)."""
        if phones <= 0 or n <= rr:
            # running-response floor is capped by the population itself
            dev = min(n, rr) if phones > 0 else 0
            return TierAllocation(tier, n, n - dev, dev, rr)
        if actors <= 0:
            return TierAllocation(tier, n, 0, n, rr)
        best = None
        # makespan is unimodal in the split; still cheap to scan exactly
        # for realistic n; for big n scan the tier at coarse + fine steps
        step = max(1, n // 4096)
        candidates = set(range(rr, n + 1, step)) | {rr, n}
        for dev in candidates:
            cost = max(_cost_logical(n - dev, actors),
                       _cost_device(dev, phones))
            if best is None or cost < best[0]:
                best = (cost, dev)
        dev = best[1]
        for d in range(max(rr, dev - step), min(n, dev + step) + 1):
            cost = max(_cost_logical(n - d, actors), _cost_device(d, phones))
            if cost < best[0]:
                best = (cost, d)
        dev = best[1]
        return TierAllocation(tier, n, n - dev, dev, rr)


def makespan(alloc: DataAllocation, actors_map: Dict[str, int],
             phones_map: Dict[str, int]) -> float:
    lg = sum(_cost_logical(t.logical, actors_map.get(t.tier, 1))
             for t in alloc.tiers)
    dv = sum(_cost_device(t.device, phones_map.get(t.tier, 1))
             for t in alloc.tiers)
    return max(lg, dv)
