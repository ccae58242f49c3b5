"""GPU-resident behaviour sampler.

Turns a gradient-house strategy (deviceflow/strategy.py JSON) into
per-round cohort masks without stalling the round loop:

- offline mask: clients whose gradient the flow schedule never forwards
  this round (arrival-curve shortfall — the online/offline/spike
  behaviour of the reference's Dispatcher flow mode,
  dispatcher.py:174-242);
- drop mask: clients trained but excluded from aggregation (the
  drop_probability / drop_amounts simulation, strategy.py:275-311).

The arrival-rate functions f(t) are integrated ON DEVICE: the expression
is evaluated with torch in place of math over the whole
[slots x 100]-point grid in one shot, so a 50k-client churn config costs
microseconds per round instead of a Python loop over slots.  The
canonical scalar math (bit-parity with the reference) stays in
deviceflow/strategy.py; tests cross-check the two.
"""

from __future__ import annotations

import json
import math
from typing import Dict, Tuple

import torch

from .strategy import AREA_CALCULATION_NUM, _dct


def _torch_eval_rate(func_string: str, t: torch.Tensor) -> torch.Tensor:
    """Evaluate an arrival-rate expression over a tensor grid.

    'math.*'/'np.*' resolve to torch equivalents; plain arithmetic works
    natively on the tensor.
    """
    from .strategy import compile_rate_expr
    ns = {"math": torch, "np": torch, "t": t, "abs": torch.abs,
          "min": torch.minimum, "max": torch.maximum, "__builtins__": {}}
    out = eval(compile_rate_expr(func_string), ns)
    if not torch.is_tensor(out):
        out = torch.full_like(t, float(out))
    return out


class BehaviorSampler:
    def __init__(self, strategy: str, seed: int = 0, device: str = "cpu",
                 task_id: str = "task", operator: str = "train"):
        self.spec = json.loads(strategy) if strategy else {}
        self.device = torch.device(device)
        self.gen = torch.Generator(device=self.device)
        self.gen.manual_seed(seed)
        self.task_id = task_id
        self.operator = operator
        self._cache: Dict[int, Tuple[float, float]] = {}

    # ------------------------------------------------------------------
    def _flow_fractions(self, round_idx: int) -> Tuple[float, float]:
        """(schedule_total, dropped_fraction_of_forwarded) for flow mode.

        schedule_total is the flow's dispatch budget
        (total_dispatch_amount): a cohort larger than it sees the
        surplus clients as offline shortfall (the dispatcher forwards
        only the scheduled amounts before round release; reference
        dispatcher.py:174-242), <= 0 meaning "no schedule" (forward
        everything)."""
        if round_idx in self._cache:
            return self._cache[round_idx]
        flow = _dct(self.spec.get("flow_dispatch"))
        total = flow.get("total_dispatch_amount", 0)
        spec = _dct(flow.get("specific_interval"))
        intervals = spec.get("intervals", [])
        if spec.get("time_type", "relative") != "relative":
            try:
                intervals = intervals[round_idx]
            except Exception:
                intervals = []
        rules = _dct(spec.get("dispatch_rules"))
        domains = rules.get("domains", [])
        functions = rules.get("functions", [])
        if not (intervals and len(intervals) == len(domains) == len(functions)):
            # invalid config: canonical Strategy returns [],[],[] — no
            # schedule, everything forwards
            self._cache[round_idx] = (0.0, 0.0)
            return self._cache[round_idx]
        # integrate every interval's curve on device in one grid pass
        per_area = []
        for interval, domain, func in zip(intervals, domains, functions):
            ilen = interval[1] - interval[0]
            npts = ilen * AREA_CALCULATION_NUM + 1
            grid = torch.linspace(float(domain[0]), float(domain[1]), npts,
                                  device=self.device)
            ys = _torch_eval_rate(func, grid)
            pieces = 0.5 * (ys[1:] + ys[:-1]) * ((domain[1] - domain[0]) /
                                                 max(1e-12, (npts - 1)))
            # reference counts only positive trapezoid pieces, and its
            # slot width is 1 s of *interval* time: rescale area to match
            scale = (ilen / max(1e-12, (domain[1] - domain[0]))
                     if domain[1] != domain[0] else 0.0)
            per_area.append(float(pieces.clamp_min(0).sum()) * scale)
        total_area = sum(per_area)
        if total_area <= 0 or total <= 0:
            # zero-area curves / zero budget: Strategy yields no
            # schedule -> everything forwards, nothing drops
            self._cache[round_idx] = (0.0, 0.0)
            return self._cache[round_idx]
        # the schedule apportions exactly `total` messages over the
        # slots; drops are a fraction of THOSE
        drop_spec = _dct(spec.get("drop_simulation"))
        drop_frac = 0.0
        if "drop_probability" in drop_spec:
            probs = drop_spec["drop_probability"]
            drop_frac = sum(p * a for p, a in zip(probs, per_area)) / total_area
        elif "drop_amounts" in drop_spec:
            drop_frac = min(1.0, sum(drop_spec["drop_amounts"]) / max(1, total))
        self._cache[round_idx] = (float(total), drop_frac)
        return self._cache[round_idx]

    def _timing_drop_fraction(self, round_idx: int) -> float:
        """Dropped / dispatched for a specific_timing schedule, from the
        SAME analysis the dispatcher runs (strategy.py
        _specific_timing) — flow-id round suffix selects per-round
        timing lists."""
        key = -1000 - round_idx
        if key in self._cache:
            return self._cache[key][1]
        import random as _random
        from .strategy import Strategy
        _, amounts, drops = Strategy.flow_strategy_analysis(
            json.dumps(self.spec),
            f"{self.task_id}_{self.operator}_{round_idx}",
            rng=_random.Random(round_idx * 9176 + 13))
        total = sum(amounts)
        frac = (sum(len(d) for d in drops) / total) if total else 0.0
        self._cache[key] = (1.0, frac)
        return frac

    # ------------------------------------------------------------------
    def __call__(self, round_idx: int, cohort: int
                 ) -> Tuple[torch.Tensor, torch.Tensor]:
        offline = torch.zeros(cohort, dtype=torch.bool, device=self.device)
        dropped = torch.zeros(cohort, dtype=torch.bool, device=self.device)
        if not self.spec:
            return offline, dropped

        # churn extension: direct offline/availability shaping per round.
        # offline_simulation: {"offline_probability": p} or
        #   {"spike_period": k, "spike_offline_fraction": f} —
        # periodic offline spikes (BASELINE config 4).
        off = _dct(self.spec.get("offline_simulation"))
        if off:
            p = float(off.get("offline_probability", 0.0))
            period = int(off.get("spike_period", 0))
            if period > 0 and round_idx % period == period - 1:
                p = max(p, float(off.get("spike_offline_fraction", 0.0)))
            if p > 0:
                offline |= (torch.rand(cohort, generator=self.gen,
                                       device=self.device) < p)

        rt = _dct(self.spec.get("real_time_dispatch"))
        if rt.get("use_strategy", False):
            p = float(_dct(rt.get("drop_simulation")).get("drop_probability", 0) or 0)
            if p > 0:
                dropped |= (torch.rand(cohort, generator=self.gen,
                                       device=self.device) < p)
            return offline, dropped

        flow = _dct(self.spec.get("flow_dispatch"))
        if flow.get("use_strategy", False) and \
                _dct(flow.get("specific_timing")).get("use", False):
            # timing mode: the exact (amounts, drop-index) schedule the
            # dispatcher would run — dropped fraction of the round's
            # dispatch total becomes the aggregation drop mask
            drop_frac = self._timing_drop_fraction(round_idx)
            if drop_frac > 0:
                dropped |= (torch.rand(cohort, generator=self.gen,
                                       device=self.device) < drop_frac)
            return offline, dropped
        if flow.get("use_strategy", False) and \
                _dct(flow.get("specific_interval")).get("use", False):
            sched_total, drop_frac = self._flow_fractions(round_idx)
            # forwarded fraction = schedule budget vs this round's
            # cohort: a budget below the cohort is offline shortfall
            # (the dispatcher never forwards more than the schedule)
            fwd_frac = 1.0 if sched_total <= 0 else \
                min(1.0, sched_total / max(1, cohort))
            if fwd_frac < 1.0:
                n_off = int(round((1.0 - fwd_frac) * cohort))
                if n_off > 0:
                    perm = torch.randperm(cohort, generator=self.gen,
                                          device=self.device)
                    offline[perm[:n_off]] = True
            if drop_frac > 0:
                dropped |= (torch.rand(cohort, generator=self.gen,
                                       device=self.device) < drop_frac)
            dropped &= ~offline
        return offline, dropped
