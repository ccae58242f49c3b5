"""End-to-end CPU round loop (BASELINE config 1: FedAvg 2-layer MLP,
10 virtual clients, plumbing path)."""

import torch

from olearning_sim_amd.engine import EngineJob, LogicalEngine


def _job(**kw):
    base = dict(task_id="t_cpu", model_name="mlp",
                model_kwargs={"in_features": 64, "hidden": 32, "num_classes": 10},
                clients=10, rounds=3, local_steps=2, batch_size=4,
                lr=0.1, device="cpu", dtype="float32", num_classes=10,
                shard_size=16, seed=7)
    base.update(kw)
    return EngineJob(**base)


def test_mlp_rounds_run_and_loss_drops():
    sink_rows = []
    eng = LogicalEngine(_job(rounds=5), result_sink=sink_rows.append)
    out = eng.run()
    assert out["rounds"] == 5
    assert out["success_total"] == 5 * 10
    assert out["failed_total"] == 0
    losses = [r["loss"] for r in sink_rows if r["loss"] is not None]
    assert losses[-1] < losses[0]  # training reduces loss on synthetic data
    # result rows have the reference logical_result shape
    row = sink_rows[0]
    tgt = row["logical_result"]["logical_result"][0]["simulation_target"]
    assert tgt["success_num"] == [10]
    assert tgt["failed_num"] == [0]
    assert row["logical_operator"] == "train"


def test_global_model_changes_each_round():
    eng = LogicalEngine(_job(rounds=1))
    before = eng.master.flat.clone()
    eng.run_round(0)
    assert not torch.allclose(before, eng.master.flat)


def test_fedprox_mu_changes_update():
    j1, j2 = _job(seed=3), _job(seed=3, prox_mu=0.5)
    e1, e2 = LogicalEngine(j1), LogicalEngine(j2)
    e1.run_round(0)
    e2.run_round(0)
    assert not torch.allclose(e1.master.flat, e2.master.flat)


def test_cohort_rotates_over_population():
    eng = LogicalEngine(_job(clients=10, cohort_size=4))
    seen = set()
    for r in range(5):
        seen.update(eng.select_cohort(r).tolist())
    assert seen == set(range(10))


def test_chunked_equals_unchunked():
    e1 = LogicalEngine(_job(seed=11, chunk_clients=3))
    e2 = LogicalEngine(_job(seed=11, chunk_clients=10))
    e1.run_round(0)
    e2.run_round(0)
    assert torch.allclose(e1.master.flat, e2.master.flat, atol=1e-5)


def test_behavior_offline_counts_failed():
    import json
    strategy = json.dumps({"offline_simulation": {"offline_probability": 1.0}})
    eng = LogicalEngine(_job(behavior_strategy=strategy, dynamic_num=100))
    rec = eng.run_round(0)
    assert rec["failed"] == 10
    assert rec["success"] == 0


def test_round_failed_when_over_tolerance():
    import json
    strategy = json.dumps({"offline_simulation": {"offline_probability": 1.0}})
    eng = LogicalEngine(_job(behavior_strategy=strategy, dynamic_num=2, rounds=5))
    out = eng.run()
    assert out["rounds"] == 1  # early stop: failed > dynamic_num
    assert out["records"][0]["round_failed"]


def test_engine_drives_deviceflow_lifecycle():
    """use_gradient_house: the engine notifies start/complete per round
    and chunk summaries flow through sorter -> dispatcher -> outbound."""
    import json
    import time as _time
    from olearning_sim_amd.deviceflow.service import DeviceFlowService
    strategy = json.dumps({"real_time_dispatch": {
        "use_strategy": True, "dispatch_batch_sizes": [1]}})
    svc = DeviceFlowService(time_scale=0.0, seed=1)
    svc.register_task("t_df", ["logical_simulation"])
    eng = LogicalEngine(_job(task_id="t_df", rounds=2, chunk_clients=5,
                             behavior_strategy=strategy),
                        deviceflow=svc)
    eng.run()
    t0 = _time.time()
    while _time.time() - t0 < 10 and not svc.check_dispatch_finished("t_df"):
        _time.sleep(0.02)
    assert svc.check_dispatch_finished("t_df")
    msgs = svc.outbound.drain()
    # 2 rounds x 2 chunks of 5 clients
    assert len(msgs) == 4
    assert {m.routing_key for m in msgs} == {"t_df_train_0", "t_df_train_1"}
    assert all(m.payload["clients"] == 5 for m in msgs)
    svc.shutdown()


def test_engine_eval_operator():
    eng = LogicalEngine(_job(rounds=2, eval_every=2))
    recs = eng.run()["records"]
    assert "eval_acc" not in recs[0]
    assert 0.0 <= recs[1]["eval_acc"] <= 1.0
    assert recs[1]["eval_loss"] > 0


def test_engine_records_perf_metrics():
    from olearning_sim_amd.perf import PerformanceManager
    pm = PerformanceManager()
    eng = LogicalEngine(_job(task_id="t_perf", rounds=3), perf=pm)
    eng.run()
    summ = pm.summary("t_perf")
    assert summ["metrics"]["round_time_s"]["count"] == 3
    assert summ["metrics"]["clients_per_s"]["last"] > 0


def test_federated_training_converges():
    """Semantic end-to-end: 25 FL rounds on the synthetic non-IID task
    must beat chance accuracy clearly (the whole pipeline learns)."""
    job = EngineJob(task_id="cv", model_name="mlp",
                    model_kwargs={"in_features": 64, "hidden": 32,
                                  "num_classes": 5},
                    clients=16, rounds=25, local_steps=2, batch_size=8,
                    lr=0.2, device="cpu", dtype="float32", num_classes=5,
                    shard_size=16, seed=9, eval_every=25, eval_batch=128)
    eng = LogicalEngine(job)
    out = eng.run()
    rec = out["records"][-1]
    assert rec["eval_acc"] > 0.35      # chance = 0.2 on 5 classes
    assert rec["eval_loss"] < 1.55     # below ln(5) ~ 1.61


def test_per_tier_success_failed_accounting():
    """Client ids map to device tiers by prefix ranges (reference
    High/Middle/Low tiers, task_manager.py logical_result vectors)."""
    import json
    strategy = json.dumps({"offline_simulation": {"offline_probability": 1.0}})
    eng = LogicalEngine(_job(
        behavior_strategy=strategy, dynamic_num=100,
        tier_counts=[("high", 6), ("low", 4)], dynamic_nums=[100, 100]))
    rec = eng.run_round(0)
    assert rec["failed_per_tier"] == [6, 4]
    assert rec["success_per_tier"] == [0, 0]


def test_per_tier_result_rows_and_tolerance():
    rows = []
    eng = LogicalEngine(_job(rounds=1,
                             tier_counts=[("high", 7), ("middle", 3)],
                             dynamic_nums=[0, 0]),
                        result_sink=rows.append)
    out = eng.run()
    rec = out["records"][0]
    assert rec["success_per_tier"] == [7, 3]
    assert not rec["round_failed"]
    tgt = rows[0]["logical_result"]["logical_result"][0]["simulation_target"]
    assert tgt["devices"] == ["high", "middle"]
    assert tgt["success_num"] == [7, 3]
    assert tgt["failed_num"] == [0, 0]


def test_per_tier_tolerance_fails_on_one_tier():
    """round_failed is per tier: a tier over its own dynamic_num fails
    the round even if the total stays under the summed tolerance."""
    import json
    strategy = json.dumps({"offline_simulation": {"offline_probability": 1.0}})
    eng = LogicalEngine(_job(behavior_strategy=strategy, rounds=1,
                             dynamic_num=100,
                             tier_counts=[("high", 6), ("low", 4)],
                             dynamic_nums=[100, 2]))
    rec = eng.run_round(0)
    assert rec["round_failed"]  # low tier: 4 failed > 2 tolerated
