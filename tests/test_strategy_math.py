"""Deviceflow behaviour-model math, checked against hand-computed values
of the reference algorithm (ols_core/deviceflow/non_grpc/strategy.py)."""

import json
import math
import random

import pytest

from olearning_sim_amd.deviceflow.strategy import Strategy


def flow(spec):
    return json.dumps({"flow_dispatch": dict({"use_strategy": True}, **spec)})


def test_real_time_detection_and_analysis():
    s = json.dumps({"real_time_dispatch": {
        "use_strategy": True, "dispatch_batch_sizes": [10, 20],
        "drop_simulation": {"drop_probability": 0.25}}})
    assert Strategy.check_real_time_dispatch(s)
    sizes, p = Strategy.real_time_strategy_analysis(s)
    assert sizes == [10, 20] and p == 0.25
    assert not Strategy.check_real_time_dispatch(json.dumps({}))


def test_specific_timing_relative():
    s = flow({"total_dispatch_amount": 60,
              "specific_timing": {"use": True, "time_type": "relative",
                                  "timings": [0, 5, 10],
                                  "amounts": [10, 20, 30]}})
    timing, amounts, drops = Strategy.flow_strategy_analysis(s, "t_op_0")
    assert timing == [0, 5, 10]
    assert amounts == [10, 20, 30]
    assert drops == [[], [], []]


def test_specific_timing_mismatched_lengths_rejected():
    s = flow({"total_dispatch_amount": 10,
              "specific_timing": {"use": True, "timings": [0, 1],
                                  "amounts": [10]}})
    assert Strategy.flow_strategy_analysis(s, "t_op_0") == ([], [], [])


def test_both_timing_and_interval_rejected():
    s = flow({"total_dispatch_amount": 10,
              "specific_timing": {"use": True, "timings": [0], "amounts": [10]},
              "specific_interval": {"use": True}})
    assert Strategy.flow_strategy_analysis(s, "t_op_0") == ([], [], [])


def test_interval_constant_rate_splits_evenly():
    # f(t)=1 over a 10 s interval: 10 slots, equal area -> 10 msgs/slot
    s = flow({"total_dispatch_amount": 100,
              "specific_interval": {
                  "use": True, "time_type": "relative",
                  "intervals": [[0, 10]],
                  "dispatch_rules": {"domains": [[0.0, 10.0]],
                                     "functions": ["1"]}}})
    timing, amounts, drops = Strategy.flow_strategy_analysis(s, "t_op_0")
    assert len(amounts) == 10
    assert sum(amounts) == 100
    assert amounts == [10] * 10
    assert timing == [0] + [1] * 9


def test_interval_sin_spike_total_preserved():
    # the reference README example: math.sin(t)+1 over [0, 6.28]
    s = flow({"total_dispatch_amount": 500,
              "specific_interval": {
                  "use": True,
                  "intervals": [[0, 10]],
                  "dispatch_rules": {"domains": [[0.0, 6.28]],
                                     "functions": ["math.sin(t)+1"]}}})
    timing, amounts, drops = Strategy.flow_strategy_analysis(s, "t_op_0")
    assert sum(amounts) == 500
    assert len(amounts) == 10
    # rate peaks near t=pi/2 (slot 2-3 of 10) and dips near 3*pi/2
    assert max(amounts) == amounts[2]
    assert min(amounts) == amounts[7]


def test_two_intervals_area_apportioning():
    # f=2 on 5 s vs f=1 on 5 s: amounts split 2:1
    s = flow({"total_dispatch_amount": 150,
              "specific_interval": {
                  "use": True,
                  "intervals": [[0, 5], [10, 15]],
                  "dispatch_rules": {"domains": [[0.0, 5.0], [0.0, 5.0]],
                                     "functions": ["2", "1"]}}})
    timing, amounts, drops = Strategy.flow_strategy_analysis(s, "t_op_0")
    assert sum(amounts) == 150
    assert sum(amounts[:5]) == 100
    assert sum(amounts[5:]) == 50
    # gap between interval 0 (slots 0..4) and interval 1 (slots 10..14)
    assert timing[5] == 6  # 10 - 4


def test_negative_rate_only_positive_area_counts():
    # f(t) = -1 everywhere -> zero positive area -> rejected
    s = flow({"total_dispatch_amount": 10,
              "specific_interval": {
                  "use": True, "intervals": [[0, 5]],
                  "dispatch_rules": {"domains": [[0.0, 5.0]],
                                     "functions": ["-1"]}}})
    assert Strategy.flow_strategy_analysis(s, "t_op_0") == ([], [], [])


def test_drop_probability_bounds():
    rng = random.Random(0)
    drops = Strategy.generate_drop_list(
        [10, 10, 10], {"drop_probability": [0, 1, 0.5]}, rng)
    assert drops[0] == []
    assert drops[1] == list(range(10))
    assert all(0 <= i < 10 for i in drops[2])


def test_drop_amounts_exact():
    rng = random.Random(1)
    drops = Strategy.generate_drop_list(
        [10, 10, 10], {"drop_amounts": [0, 4, 15]}, rng)
    assert drops[0] == []
    assert len(drops[1]) == 4 and drops[1] == sorted(drops[1])
    assert drops[2] == list(range(10))   # over-drop clamps to everything


def test_interval_drop_probability_expansion():
    s = flow({"total_dispatch_amount": 100,
              "specific_interval": {
                  "use": True, "intervals": [[0, 10]],
                  "dispatch_rules": {"domains": [[0.0, 10.0]],
                                     "functions": ["1"]},
                  "drop_simulation": {"drop_probability": [1.0]}}})
    timing, amounts, drops = Strategy.flow_strategy_analysis(
        s, "t_op_0", rng=random.Random(3))
    assert all(d == list(range(a)) for d, a in zip(drops, amounts))


def test_round_indexed_absolute_interval_uses_flow_round():
    base = {"total_dispatch_amount": 50,
            "specific_interval": {
                "use": True, "time_type": "absolute",
                "intervals": [
                    [["2099-01-01 00:00:00", "2099-01-01 00:00:05"]],
                    [["2099-01-01 01:00:00", "2099-01-01 01:00:05"]]],
                "dispatch_rules": {"domains": [[0.0, 5.0]],
                                   "functions": ["1"]}}}
    from datetime import datetime
    now = datetime(2098, 12, 31, 23, 59, 0)
    t0, a0, _ = Strategy.flow_strategy_analysis(flow(base), "t_op_0", now=now)
    t1, a1, _ = Strategy.flow_strategy_analysis(flow(base), "t_op_1", now=now)
    assert sum(a0) == 50 and sum(a1) == 50
    # round 1 starts one hour later than round 0
    assert t1[0] - t0[0] == pytest.approx(3600, abs=1)


def test_sampler_drop_fraction_matches_strategy_schedule():
    """The GPU-resident sampler's flow fractions must agree with the
    canonical CPU schedule math on the same strategy JSON."""
    import torch
    from olearning_sim_amd.deviceflow.sampler import BehaviorSampler
    s = flow({"total_dispatch_amount": 1000,
              "specific_interval": {
                  "use": True, "intervals": [[0, 10]],
                  "dispatch_rules": {"domains": [[0.0, 6.28]],
                                     "functions": ["math.sin(t)+1"]},
                  "drop_simulation": {"drop_probability": [0.3]}}})
    # canonical: full schedule forwards everything minus the drops
    timing, amounts, drops = Strategy.flow_strategy_analysis(
        s, "t_train_0", rng=random.Random(0))
    assert sum(amounts) == 1000
    canonical_drop = sum(len(d) for d in drops) / 1000
    sampler = BehaviorSampler(s, seed=0, device="cpu")
    sched_total, drop_frac = sampler._flow_fractions(0)
    # the sampler sees the dispatcher's schedule budget (sum(amounts))
    assert sched_total == sum(amounts) == 1000
    assert abs(drop_frac - 0.3) < 0.02          # configured probability
    assert abs(canonical_drop - drop_frac) < 0.06  # sampled realisation
    # cohort within the schedule budget: nobody offline, drops ~ p
    offline, dropped = sampler(0, 1000)
    assert int(offline.sum()) == 0
    assert abs(float(dropped.float().mean()) - 0.3) < 0.04


def test_sampler_interval_shortfall_is_offline():
    """A flow schedule whose total_dispatch_amount is below the cohort
    forwards only the scheduled amount: the surplus clients are offline
    shortfall (dispatcher forwards exactly the schedule's amounts;
    reference dispatcher.py:174-242), and drops apply to forwarded
    clients only."""
    from olearning_sim_amd.deviceflow.sampler import BehaviorSampler
    s = flow({"total_dispatch_amount": 1000,
              "specific_interval": {
                  "use": True, "intervals": [[0, 10]],
                  "dispatch_rules": {"domains": [[0.0, 6.28]],
                                     "functions": ["math.sin(t)+1"]},
                  "drop_simulation": {"drop_probability": [0.3]}}})
    sampler = BehaviorSampler(s, seed=0, device="cpu")
    offline, dropped = sampler(0, 4000)
    assert int(offline.sum()) == 3000           # 4000 cohort - 1000 budget
    assert not bool((dropped & offline).any())  # drops only among online
    online_drop = float(dropped.sum()) / 1000
    assert abs(online_drop - 0.3) < 0.06


def test_sampler_torch_integration_matches_scalar_area():
    """Device-side trapezoid integration == the scalar math's area."""
    import math as m
    from olearning_sim_amd.deviceflow.sampler import _torch_eval_rate
    import torch
    grid = torch.linspace(0.0, 6.28, 1001)
    ys = _torch_eval_rate("math.sin(t)+1", grid)
    area_t = float((0.5 * (ys[1:] + ys[:-1]) * (6.28 / 1000)).sum())
    # analytic: integral of sin+1 over [0, 6.28]
    area_ref = (-m.cos(6.28) + m.cos(0.0)) + 6.28
    assert abs(area_t - area_ref) < 1e-3


def test_sampler_timing_mode_drop_fraction():
    """specific_timing strategies map to drop masks with the dispatch
    schedule's exact dropped/dispatched ratio."""
    import torch
    from olearning_sim_amd.deviceflow.sampler import BehaviorSampler
    spec = {"flow_dispatch": {
        "use_strategy": True, "total_dispatch_amount": 100,
        "specific_timing": {
            "use": True,
            "timings": [0, 1, 2, 3],
            "amounts": [25, 25, 25, 25],
            "drop_simulation": {"drop_amounts": [5, 5, 5, 5]}}}}
    s = BehaviorSampler(json.dumps(spec), seed=3, device="cpu")
    assert abs(s._timing_drop_fraction(0) - 0.2) < 1e-9
    off, drop = s(0, 1000)
    assert not off.any()
    # ~20% dropped (binomial around 200 of 1000)
    n = int(drop.sum())
    assert 120 <= n <= 280


def test_sampler_timing_mode_no_drops_is_noop():
    import torch
    from olearning_sim_amd.deviceflow.sampler import BehaviorSampler
    spec = {"flow_dispatch": {
        "use_strategy": True, "total_dispatch_amount": 10,
        "specific_timing": {"use": True, "timings": [0, 1],
                            "amounts": [5, 5]}}}
    s = BehaviorSampler(json.dumps(spec), seed=3, device="cpu")
    off, drop = s(1, 64)
    assert not off.any() and not drop.any()


def test_rate_expression_whitelist():
    """Arrival-rate expressions are AST-whitelisted: arithmetic and
    math.*/np.* only — attribute-chain escapes and arbitrary names are
    rejected at parse time (the reference bare-evals task input)."""
    from olearning_sim_amd.deviceflow.strategy import (_eval_rate,
                                                       RateExprError)
    import math as m
    assert _eval_rate("math.sin(t)+1", 0.3) == pytest.approx(m.sin(0.3) + 1)
    assert _eval_rate("min(2.0, max(t, 0.5)) * 2", 0.1) == pytest.approx(1.0)
    assert _eval_rate("np.exp(-t)", 1.0) == pytest.approx(m.exp(-1.0))
    for bad in (
        "abs.__self__",                       # builtins escape
        "().__class__",                       # object traversal
        "__import__('os')",                   # unknown name
        "math.__dict__",                      # dunder attribute
        "[x for x in (1,)]",                  # comprehension
        "'a'*9999999",                        # non-numeric literal
        "lambda: 1",
    ):
        with pytest.raises(RateExprError):
            _eval_rate(bad, 0.0)
