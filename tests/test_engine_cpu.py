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
