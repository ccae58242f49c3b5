"""Client-batched model semantics: every client's forward must equal an
independent per-client computation (the batching is an implementation
detail, never a semantic change)."""

import torch

from olearning_sim_amd.models import build_model
from olearning_sim_amd.engine.client_manager import (
    FlatParams, replicate_flat, batched_views)


def _per_client_reference(model, params_batched, x):
    """Run each client alone (C=1) and stack."""
    outs = []
    C = x.shape[0]
    for c in range(C):
        p1 = {k: v[c:c + 1] for k, v in params_batched.items()}
        outs.append(model.forward(p1, x[c:c + 1]))
    return torch.cat(outs, dim=0)


def _check_model(model, B=3, C=4, atol=1e-5):
    gen = torch.Generator().manual_seed(0)
    gp = model.init_global(generator=gen)
    master = FlatParams(gp)
    cast = master.cast(torch.float32)
    buf = replicate_flat(cast, C)
    with torch.no_grad():
        # perturb so clients differ
        buf += 0.01 * torch.randn(buf.shape, generator=gen)
    params = batched_views(buf.detach(), master.shapes, C)
    x = torch.randn((C, B) + model.input_shape, generator=gen)
    got = model.forward(params, x)
    want = _per_client_reference(model, params, x)
    assert got.shape == (C, B, model.num_classes)
    torch.testing.assert_close(got, want, atol=atol, rtol=1e-4)


def test_mlp_batching_matches_per_client():
    _check_model(build_model("mlp", in_features=32, hidden=16, num_classes=5))


def test_lenet_batching_matches_per_client():
    _check_model(build_model("lenet", num_classes=10))


def test_resnet18_batching_matches_per_client():
    _check_model(build_model("resnet18", num_classes=10, width_mult=0.25),
                 B=2, C=3, atol=1e-4)


def test_resnet18_full_width_param_count():
    m = build_model("resnet18", num_classes=100)
    gp = m.init_global()
    total = sum(v.numel() for v in gp.values())
    # torchvision resnet18 has 11.69M params at 1000 classes; the CIFAR
    # stem (3x3, no maxpool) + GN + 100 classes lands close to 11.2M
    assert 10_500_000 < total < 11_800_000


def test_backward_produces_grads_for_all_params():
    model = build_model("mlp", in_features=16, hidden=8, num_classes=4)
    gp = model.init_global(generator=torch.Generator().manual_seed(1))
    master = FlatParams(gp)
    buf = replicate_flat(master.cast(torch.float32), 2)
    params = batched_views(buf, master.shapes, 2)
    x = torch.randn(2, 3, 16)
    y = torch.randint(0, 4, (2, 3))
    loss = model.loss(params, x, y)
    g, = torch.autograd.grad(loss, [buf])
    assert g.shape == buf.shape
    assert float(g.abs().sum()) > 0


def test_bert_tiny_batching_matches_per_client():
    m = build_model("bert", seq_len=8, vocab_size=50, hidden=32, layers=2,
                    heads=2)
    gen = torch.Generator().manual_seed(0)
    gp = m.init_global(generator=gen)
    master = FlatParams(gp)
    C, B = 3, 2
    buf = replicate_flat(master.cast(torch.float32), C)
    with torch.no_grad():
        buf += 0.01 * torch.randn(buf.shape, generator=gen)
    params = batched_views(buf.detach(), master.shapes, C)
    x = torch.randint(0, 50, (C, B, 8), generator=gen)
    got = m.forward(params, x)
    want = torch.cat([m.forward({k: v[c:c + 1] for k, v in params.items()},
                                x[c:c + 1]) for c in range(C)])
    assert got.shape == (C, B, 8, 50)
    torch.testing.assert_close(got, want, atol=1e-4, rtol=1e-4)


def test_bert_engine_round_cpu():
    from olearning_sim_amd.engine import EngineJob, LogicalEngine
    job = EngineJob(task_id="b", model_name="bert",
                    model_kwargs={"seq_len": 8, "vocab_size": 50,
                                  "hidden": 32, "layers": 2, "heads": 2},
                    clients=4, rounds=2, local_steps=1, batch_size=2,
                    lr=0.05, device="cpu", dtype="float32",
                    vocab_size=50, seq_len=8, seed=3)
    rows = []
    eng = LogicalEngine(job, result_sink=rows.append)
    out = eng.run()
    assert out["rounds"] == 2 and out["success_total"] == 8
    assert eng.master.flat.isfinite().all()


def test_lstm_batching_matches_per_client():
    m = build_model("lstm", vocab_size=30, embed=8, hidden=16, layers=2,
                    seq_len=6)
    gen = torch.Generator().manual_seed(0)
    gp = m.init_global(generator=gen)
    master = FlatParams(gp)
    C, B = 3, 2
    buf = replicate_flat(master.cast(torch.float32), C)
    with torch.no_grad():
        buf += 0.01 * torch.randn(buf.shape, generator=gen)
    params = batched_views(buf.detach(), master.shapes, C)
    x = torch.randint(0, 30, (C, B, 6), generator=gen)
    got = m.forward(params, x)
    want = torch.cat([m.forward({k: v[c:c + 1] for k, v in params.items()},
                                x[c:c + 1]) for c in range(C)])
    assert got.shape == (C, B, 6, 30)
    torch.testing.assert_close(got, want, atol=1e-5, rtol=1e-4)


def test_lstm_param_shapes_match_init():
    m = build_model("lstm", vocab_size=30, embed=8, hidden=16, layers=2)
    gp = m.init_global()
    assert {k: tuple(v.shape) for k, v in gp.items()} == m.param_shapes()


def test_lstm_engine_round_cpu():
    from olearning_sim_amd.engine import EngineJob, LogicalEngine
    job = EngineJob(task_id="l", model_name="lstm",
                    model_kwargs={"vocab_size": 30, "embed": 8,
                                  "hidden": 16, "layers": 2, "seq_len": 6},
                    clients=4, rounds=3, local_steps=1, batch_size=2,
                    lr=0.5, device="cpu", dtype="float32",
                    vocab_size=30, seq_len=6, seed=3)
    rows = []
    eng = LogicalEngine(job, result_sink=rows.append)
    out = eng.run()
    assert out["rounds"] == 3 and out["success_total"] == 12
    assert eng.master.flat.isfinite().all()
    losses = [r["loss"] for r in rows if r["loss"] is not None]
    assert losses[-1] < losses[0]


def test_lstm_chunked_equals_unchunked():
    from olearning_sim_amd.engine import EngineJob, LogicalEngine
    def job(chunk):
        return EngineJob(task_id="lc", model_name="lstm",
                         model_kwargs={"vocab_size": 20, "embed": 4,
                                       "hidden": 8, "layers": 1,
                                       "seq_len": 5},
                         clients=6, rounds=1, local_steps=1, batch_size=2,
                         lr=0.2, device="cpu", dtype="float32",
                         vocab_size=20, seq_len=5, seed=13,
                         chunk_clients=chunk)
    e1, e2 = LogicalEngine(job(2)), LogicalEngine(job(6))
    e1.run_round(0)
    e2.run_round(0)
    torch.testing.assert_close(e1.master.flat, e2.master.flat, atol=1e-5,
                               rtol=1e-5)


def test_lstm_evaluate_global_runs():
    from olearning_sim_amd.engine import EngineJob, LogicalEngine
    job = EngineJob(task_id="le", model_name="lstm",
                    model_kwargs={"vocab_size": 20, "embed": 4,
                                  "hidden": 8, "layers": 1, "seq_len": 5},
                    clients=4, rounds=1, local_steps=1, batch_size=2,
                    lr=0.2, device="cpu", dtype="float32",
                    vocab_size=20, seq_len=5, seed=13)
    out = LogicalEngine(job).evaluate_global(0)
    assert 0.0 <= out["eval_acc"] <= 1.0
    assert out["eval_loss"] > 0


def test_resnet_custom_conv_path_matches_default_cpu():
    """forward_cbf (the hand-written-kernel path; CPU fallbacks here)
    must equal the grouped-conv default path."""
    m = build_model("resnet18", num_classes=10, width_mult=0.25)
    gen = torch.Generator().manual_seed(4)
    gp = m.init_global(generator=gen)
    master = FlatParams(gp)
    C, B = 2, 2
    buf = replicate_flat(master.cast(torch.float32), C)
    with torch.no_grad():
        buf += 0.01 * torch.randn(buf.shape, generator=gen)
    params = batched_views(buf.detach(), master.shapes, C)
    x = torch.randn((C, B) + m.input_shape, generator=gen)
    default = m.forward(params, x)
    custom = m.forward_cbf(params, x)
    torch.testing.assert_close(custom, default, atol=2e-4, rtol=1e-3)


def test_resnet_custom_conv_path_backward_matches_default_cpu():
    m = build_model("resnet18", num_classes=10, width_mult=0.25)
    gen = torch.Generator().manual_seed(5)
    gp = m.init_global(generator=gen)
    master = FlatParams(gp)
    C, B = 2, 2
    x = torch.randn((C, B) + m.input_shape, generator=gen)
    y = torch.randint(0, 10, (C, B), generator=gen)

    def grad_of(fwd):
        buf = replicate_flat(master.cast(torch.float32), C)
        params = batched_views(buf, master.shapes, C)
        logits = fwd(params, x)
        loss = torch.nn.functional.cross_entropy(
            logits.reshape(C * B, -1), y.reshape(-1))
        g, = torch.autograd.grad(loss, [buf])
        return g

    g_default = grad_of(m.forward)
    g_custom = grad_of(m.forward_cbf)
    torch.testing.assert_close(g_custom, g_default, atol=5e-4, rtol=5e-3)


def test_bert_tied_head_ce_matches_full_logits():
    import pytest
    """Sliced tied-head+CE (no logits materialisation) == the full
    logits + cross-entropy path: loss and all gradients."""
    import torch
    from olearning_sim_amd.models.bert import BertTiny, tied_head_ce
    from olearning_sim_amd.models.bert import _TiedHeadCE
    m = BertTiny(seq_len=8, vocab_size=50, hidden=16, layers=1, heads=2)
    g = torch.Generator().manual_seed(0)
    C, B, L = 3, 2, 8
    H, V = 16, 50
    hs0 = torch.randn(C, B * L, H, generator=g)
    tok0 = torch.randn(C, V, H, generator=g) * 0.2
    bias0 = torch.randn(C, V, generator=g) * 0.1
    y = torch.randint(0, V, (C, B * L), generator=g)

    old_slice = _TiedHeadCE.SLICE_V
    _TiedHeadCE.SLICE_V = 16           # force several slices + tail
    try:
        hs = hs0.clone().requires_grad_(True)
        tok = tok0.clone().requires_grad_(True)
        bias = bias0.clone().requires_grad_(True)
        loss = tied_head_ce(hs, tok, bias, y)
        loss.backward()

        hs_r = hs0.clone().requires_grad_(True)
        tok_r = tok0.clone().requires_grad_(True)
        bias_r = bias0.clone().requires_grad_(True)
        logits = torch.bmm(hs_r, tok_r.transpose(1, 2)) + bias_r.unsqueeze(1)
        ref = torch.nn.functional.cross_entropy(
            logits.reshape(C * B * L, V), y.reshape(-1))
        ref.backward()

        assert float(loss) == pytest.approx(float(ref), rel=1e-5)
        torch.testing.assert_close(hs.grad, hs_r.grad, atol=1e-5, rtol=1e-4)
        torch.testing.assert_close(tok.grad, tok_r.grad, atol=1e-5, rtol=1e-4)
        torch.testing.assert_close(bias.grad, bias_r.grad, atol=1e-5,
                                   rtol=1e-4)
    finally:
        _TiedHeadCE.SLICE_V = old_slice
