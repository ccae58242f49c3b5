"""Property-based robustness tests (hypothesis): the public parsers and
validators must never crash on arbitrary input — they return False /
empty schedules instead (the reference's swallow-and-log policy)."""

import json

from hypothesis import given, settings, strategies as st

from olearning_sim_amd.deviceflow.strategy import Strategy
from olearning_sim_amd.deviceflow.validate import ValidateStrategy
from olearning_sim_amd.task.schema import json2taskconfig, taskconfig2json
from olearning_sim_amd.task.validate import ValidateParameters

json_scalars = st.one_of(st.none(), st.booleans(),
                         st.integers(-10**6, 10**6),
                         st.floats(allow_nan=False, allow_infinity=False,
                                   width=32),
                         st.text(max_size=20))
json_values = st.recursive(
    json_scalars,
    lambda children: st.one_of(
        st.lists(children, max_size=4),
        st.dictionaries(st.text(max_size=12), children, max_size=4)),
    max_leaves=12)
json_objects = st.dictionaries(st.text(max_size=16), json_values, max_size=6)


@settings(max_examples=150, deadline=None)
@given(json_objects)
def test_task_parsing_and_validation_never_crash(raw):
    """json2taskconfig on structurally arbitrary dicts either parses or
    raises cleanly; validation returns a bool, never raises."""
    text = json.dumps(raw)
    try:
        cfg = json2taskconfig(text)
    except (TypeError, AttributeError, ValueError, KeyError):
        return   # malformed section types rejected at parse: acceptable
    v = ValidateParameters()
    ok = v.validate_task_parameters(raw, cfg)
    assert isinstance(ok, bool)
    if ok:
        # anything accepted must round-trip
        json.loads(taskconfig2json(cfg))


@settings(max_examples=150, deadline=None)
@given(json_objects)
def test_strategy_validation_never_crashes(raw):
    v = ValidateStrategy()
    assert isinstance(v.check(json.dumps(raw)), bool)


@settings(max_examples=100, deadline=None)
@given(json_objects)
def test_flow_analysis_never_crashes_on_garbage(raw):
    out = Strategy.flow_strategy_analysis(json.dumps(raw), "t_op_0")
    assert isinstance(out, tuple) and len(out) == 3


@settings(max_examples=60, deadline=None)
@given(total=st.integers(1, 5000),
       length=st.integers(1, 30),
       d0=st.floats(0, 5, allow_nan=False),
       dlen=st.floats(0.1, 10, allow_nan=False))
def test_interval_schedule_preserves_total(total, length, d0, dlen):
    """Whatever the interval/domain geometry, a valid positive-rate
    schedule apportions exactly total_dispatch_amount."""
    s = json.dumps({"flow_dispatch": {
        "use_strategy": True, "total_dispatch_amount": total,
        "specific_interval": {
            "use": True, "intervals": [[0, length]],
            "dispatch_rules": {"domains": [[d0, d0 + dlen]],
                               "functions": ["t*0 + 1"]}}}})
    timing, amounts, drops = Strategy.flow_strategy_analysis(s, "t_op_0")
    assert sum(amounts) == total
    assert len(amounts) == length
    assert all(a >= 0 for a in amounts)
    assert len(drops) == length


@settings(max_examples=60, deadline=None)
@given(st.lists(st.integers(0, 50), min_size=1, max_size=20),
       st.floats(0, 1, allow_nan=False))
def test_drop_list_bounds(amounts, p):
    drops = Strategy.generate_drop_list(
        amounts, {"drop_probability": [p] * len(amounts)})
    assert len(drops) == len(amounts)
    for d, a in zip(drops, amounts):
        assert all(0 <= i < max(1, a) for i in d)
        assert len(d) <= a


@settings(max_examples=120, deadline=None)
@given(n=st.integers(0, 5000), actors=st.integers(0, 64),
       phones=st.integers(0, 50), rr=st.integers(0, 100))
def test_auto_allocation_partitions_population(n, actors, phones, rr):
    """The hybrid optimizer always partitions each tier's population
    exactly between the two sides and honours the running-response
    floor (utils_runner.py:939-1022 cost model)."""
    from olearning_sim_amd.task.allocation import HybridOptimizer
    t = HybridOptimizer._optimize_tier("high", n, actors, phones, rr)
    assert t.logical >= 0 and t.device >= 0
    assert t.logical + t.device == n
    if phones > 0 and n > rr:
        assert t.device >= rr


@settings(max_examples=60, deadline=None)
@given(log=st.lists(st.integers(0, 500), min_size=1, max_size=3),
       dev=st.lists(st.integers(0, 500), min_size=1, max_size=3),
       dyn=st.lists(st.integers(0, 50), min_size=1, max_size=3))
def test_submitter_sides_partition_any_split(log, dev, dyn):
    """assemble_info_* always partitions nums, and the two sides'
    scaled dynamic_nums never exceed the original tolerance."""
    import copy
    from test_schema import EXAMPLE
    from olearning_sim_amd.task import json2taskconfig as j2t
    from olearning_sim_amd.task.allocation import HybridOptimizer
    from olearning_sim_amd.task.submitter import JobSubmitter
    k = min(len(log), len(dev), len(dyn))
    log, dev, dyn = log[:k], dev[:k], dyn[:k]
    nums = [a + b for a, b in zip(log, dev)]
    if sum(nums) == 0:
        return
    raw = copy.deepcopy(EXAMPLE)
    d = raw["target"]["data"][0]
    tiers = [f"t{i}" for i in range(k)]
    d["total_simulation"] = {"devices": tiers, "nums": nums,
                             "dynamic_nums": dyn}
    d["allocation"] = {"optimization": False, "logical_simulation": log,
                       "device_simulation": dev,
                       "running_response": {"devices": [], "nums": []}}
    task = j2t(json.dumps(raw))
    sub = JobSubmitter(task, HybridOptimizer(task).allocate())
    ls = sub.assemble_info_logical_simulation()
    ds = sub.assemble_info_device_simulation()
    got_l = (ls["target"]["data"][0]["total_simulation"]
             if ls else {"nums": [], "dynamic_nums": []})
    got_d = (ds["target"]["data"][0]["total_simulation"]
             if ds else {"nums": [], "dynamic_nums": []})
    # per-tier partition of nums (absent side contributes 0)
    for i, n in enumerate(nums):
        ln = got_l["nums"][i] if i < len(got_l["nums"]) else 0
        dn = got_d["nums"][i] if i < len(got_d["nums"]) else 0
        if ls and ds:
            assert ln + dn == n
    assert (sum(got_l.get("dynamic_nums", []))
            + sum(got_d.get("dynamic_nums", []))) <= sum(dyn)


@settings(max_examples=80, deadline=None)
@given(bounds=st.lists(st.integers(1, 50), min_size=1, max_size=5),
       ranges=st.lists(st.tuples(st.integers(0, 200), st.integers(0, 60)),
                       max_size=8))
def test_per_tier_counts_conserve_population(bounds, ranges):
    """Segment intersection accounting: success+failed == tier size per
    tier, and totals never go negative, for any failed ranges."""
    from olearning_sim_amd.engine.script_op import per_tier_counts
    tier_bounds = [0]
    for b in bounds:
        tier_bounds.append(tier_bounds[-1] + b)
    # clip ranges into [0, clients) and make them well-formed
    clients = tier_bounds[-1]
    fr = []
    seen = set()
    for lo, ln in ranges:
        lo = lo % clients
        hi = min(clients, lo + ln)
        if lo < hi and all(hi <= a or lo >= b for a, b in seen):
            fr.append((lo, hi))
            seen.add((lo, hi))
    succ, fail = per_tier_counts(fr, tier_bounds)
    total_failed = sum(hi - lo for lo, hi in fr)
    assert sum(fail) == total_failed
    for t in range(len(bounds)):
        assert succ[t] + fail[t] == bounds[t]
        assert succ[t] >= 0 and fail[t] >= 0


@settings(max_examples=100, deadline=None)
@given(succ=st.lists(st.integers(0, 20), min_size=1, max_size=3),
       fail=st.lists(st.integers(0, 20), min_size=1, max_size=3),
       nums=st.lists(st.integers(1, 20), min_size=1, max_size=3),
       dyn=st.lists(st.integers(0, 5), min_size=1, max_size=3),
       rnd=st.integers(0, 3))
def test_status_fusion_never_crashes_and_is_sound(succ, fail, nums, dyn, rnd):
    """Fusion accepts any result-vector shape without crashing, and
    SUCCEEDED implies every tier met its tolerance."""
    import sys
    sys.path.insert(0, "tests")
    from test_status_fusion import mk
    from olearning_sim_amd.task.status import TaskStatus
    k = len(nums)
    dyn = (dyn + [0] * k)[:k]
    mgr, table = mk()
    table.add_task("tf")
    total = {"max_round": 2, "operator_name_list": ["train"],
             "data_name_list": ["d0"],
             "total_simulation": [{"name": "d0", "simulation_target": {
                 "devices": [f"t{i}" for i in range(k)], "nums": nums,
                 "dynamic_nums": dyn}}]}
    table.set_items("tf", total_simulation=json.dumps(total))
    table.set_item_value("tf", "logical_target", json.dumps(
        {"logical_target": [{"name": "d0", "simulation_target": {
            "devices": [f"t{i}" for i in range(k)], "nums": nums}}]}))
    table.set_items("tf", logical_result=json.dumps({"logical_result": [
        {"name": "d0", "simulation_target": {
            "devices": [f"t{i}" for i in range(len(succ))],
            "success_num": succ, "failed_num": fail}}]}),
        logical_round=rnd, logical_operator="train")
    st = mgr.combine_task_status("tf", TaskStatus.SUCCEEDED,
                                 {"is_finished": True})
    assert isinstance(st, TaskStatus)
    if st == TaskStatus.SUCCEEDED:
        assert rnd >= 2
        padded = list(succ) + [0] * max(0, k - len(succ))
        assert all(padded[i] >= nums[i] - dyn[i] for i in range(k))


@settings(max_examples=40, deadline=None)
@given(clients=st.integers(2, 20), batch=st.integers(1, 6),
       rnd=st.integers(0, 3), step=st.integers(0, 2),
       vocab=st.sampled_from([0, 17]))
def test_synthetic_data_chunk_invariance(clients, batch, rnd, step, vocab):
    """A client's batch is identical whether fetched alone or inside
    any chunk — the invariant that makes chunked == unchunked training
    exact (engine/data.py per-(seed,round,step) generators)."""
    import torch
    from olearning_sim_amd.engine.data import SyntheticFederatedData
    data = SyntheticFederatedData(
        clients=clients, num_classes=5, input_shape=(12,),
        dirichlet_alpha=0.5, shard_size=8, seed=3, device="cpu",
        vocab_size=vocab, seq_len=4 if vocab else 0)
    ids = torch.arange(clients)
    x_all, y_all = data.batch(ids, rnd, step, batch, torch.float32)
    c = clients // 2
    x_one, y_one = data.batch(ids[c:c + 1], rnd, step, batch, torch.float32)
    torch.testing.assert_close(x_all[c:c + 1], x_one)
    torch.testing.assert_close(y_all[c:c + 1], y_one)


@settings(max_examples=40, deadline=None)
@given(rounds=st.lists(st.integers(0, 40), min_size=1, max_size=6,
                       unique=True),
       prefix=st.text(alphabet=st.characters(
           whitelist_categories=("Ll", "Nd")), min_size=1, max_size=8))
def test_checkpoint_latest_round_roundtrip(tmp_path_factory, rounds, prefix):
    """save_checkpoint -> latest_round finds the max round for any
    templated style and task id."""
    import torch
    from olearning_sim_amd.engine.checkpoint import (latest_round,
                                                     save_checkpoint)
    d = str(tmp_path_factory.mktemp("ck"))
    style = prefix + "_{task_id}_{current_round}_result_model.safetensors"
    for r in rounds:
        save_checkpoint(d, "tk", r, {"w": torch.zeros(2)}, style)
    assert latest_round(d, "tk", style) == max(rounds)


@settings(max_examples=60, deadline=None)
@given(st.data())
def test_schema_roundtrip_idempotent(data):
    """taskconfig2json(json2taskconfig(x)) is a fixed point: a second
    round-trip yields byte-identical JSON."""
    import copy
    from test_schema import EXAMPLE
    raw = copy.deepcopy(EXAMPLE)
    raw["target"]["priority"] = data.draw(st.integers(0, 10))
    raw["task_id"] = "t_" + data.draw(st.text(
        alphabet="abcdefghij0123456789", min_size=1, max_size=10))
    once = taskconfig2json(json2taskconfig(json.dumps(raw)))
    twice = taskconfig2json(json2taskconfig(once))
    assert once == twice


@settings(max_examples=200, deadline=None)
@given(st.dictionaries(
    st.sampled_from(["flow_dispatch", "real_time_dispatch",
                     "offline_simulation", "specific_interval",
                     "drop_simulation", "dispatch_rules", "use_strategy",
                     "total_dispatch_amount", "intervals", "domains",
                     "functions", "drop_probability", "offline_probability"]),
    json_values, max_size=5),
    st.integers(0, 3), st.integers(1, 16))
def test_sampler_tolerates_real_key_garbage(raw, rnd, cohort):
    """BehaviorSampler with REAL schema keys holding garbage values
    (null sections, mistyped lists) never crashes."""
    from olearning_sim_amd.deviceflow.sampler import BehaviorSampler
    try:
        s = BehaviorSampler(json.dumps(raw), seed=1, device="cpu")
    except (ValueError, TypeError, KeyError):
        return
    off, drop = s(rnd, cohort)
    assert off.shape[0] == cohort and drop.shape[0] == cohort


@settings(max_examples=250, deadline=None)
@given(st.dictionaries(
    st.sampled_from(["flow_dispatch", "real_time_dispatch", "use_strategy",
                     "specific_timing", "specific_interval", "timings",
                     "amounts", "intervals", "domains", "functions",
                     "dispatch_rules", "drop_simulation", "drop_probability",
                     "drop_amounts", "total_dispatch_amount", "time_type",
                     "dispatch_batch_sizes", "use"]),
    json_values, max_size=6))
def test_strategy_tolerates_real_key_garbage(raw):
    """Strategy analysers and the validator with REAL schema keys
    holding garbage values never crash."""
    text = json.dumps(raw)
    assert isinstance(ValidateStrategy().check(text), bool)
    out = Strategy.flow_strategy_analysis(text, "t_op_0")
    assert isinstance(out, tuple) and len(out) == 3
    Strategy.check_real_time_dispatch(text)
    Strategy.real_time_strategy_analysis(text)


_TASK_KEYS = [
    "user_id", "task_id", "target", "priority", "data", "name",
    "data_path", "total_simulation", "devices", "nums", "dynamic_nums",
    "allocation", "optimization", "logical_simulation",
    "device_simulation", "running_response", "operatorflow",
    "flow_setting", "round", "start", "stop", "operators", "model",
    "use_model", "computation_unit", "setting", "num_cpus",
    "resource_request", "num_request", "operator_params",
    "operator_code_path", "operator_entry_file"]
_task_vals = st.recursive(
    st.one_of(st.none(), st.booleans(), st.integers(-100, 1000),
              st.sampled_from(["", "x", "FILE", "high", "{}", "not json"])),
    lambda ch: st.one_of(
        st.lists(ch, max_size=3),
        st.dictionaries(st.sampled_from(_TASK_KEYS), ch, max_size=5)),
    max_leaves=20)


@settings(max_examples=150, deadline=None)
@given(st.dictionaries(st.sampled_from(_TASK_KEYS), _task_vals, max_size=6))
def test_task_parsing_tolerates_real_key_garbage(raw):
    """Real schema keys holding null/mistyped values: parse either
    raises cleanly (submit wraps it) or validates to a bool, and
    anything accepted round-trips."""
    text = json.dumps(raw)
    try:
        cfg = json2taskconfig(text)
    except (TypeError, AttributeError, ValueError, KeyError):
        return
    v = ValidateParameters()
    ok = v.validate_task_parameters(raw, cfg)
    assert isinstance(ok, bool)
    if ok:
        json.loads(taskconfig2json(cfg))
