"""Numerics of the HIP kernels vs plain fp32 PyTorch references.
All tests require the MI355X box (pytest -m gpu)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _require_ext():
    from olearning_sim_amd.ops import load_hip_ops
    load_hip_ops(required=True)


def _layout(shapes, clients, dtype, seed=0):
    """Build (buf, grad, master, offsets) in the engine's param-major layout."""
    g = torch.Generator().manual_seed(seed)
    numels = [int(torch.tensor(s).prod()) for s in shapes]
    P = sum(numels)
    master = torch.randn(P, generator=g).to(dtype).cuda()
    buf = torch.randn(clients * P, generator=g).to(dtype).cuda()
    grad = torch.randn(clients * P, generator=g).to(dtype).cuda()
    offs = torch.tensor([0] + list(torch.tensor(numels).cumsum(0)),
                        dtype=torch.int64).cuda()
    return buf, grad, master, offs


SHAPES = [(64, 3, 3, 3), (128,), (100, 512), (7,)]


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_fused_sgd_plain(dtype):
    from olearning_sim_amd.ops import fused
    buf, grad, master, offs = _layout(SHAPES, 5, dtype)
    want = (buf.float() - 0.1 * grad.float())
    fused.fused_sgd_update_flat(buf, grad, master, 5, lr=0.1, mu=0.0,
                                offsets=offs)
    tol = 1e-6 if dtype == torch.float32 else 4e-2
    torch.testing.assert_close(buf.float(), want.to(dtype).float(),
                               atol=tol, rtol=tol)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_fused_sgd_prox_matches_cpu_reference(dtype):
    from olearning_sim_amd.ops import fused
    C = 5
    buf, grad, master, offs = _layout(SHAPES, C, dtype)
    ref = buf.clone().cpu()
    gr = grad.clone().cpu()
    fused.fused_sgd_update_flat(ref, gr, master.cpu(), C, lr=0.1, mu=0.3,
                                offsets=offs.cpu())
    fused.fused_sgd_update_flat(buf, grad, master, C, lr=0.1, mu=0.3,
                                offsets=offs)
    tol = 1e-6 if dtype == torch.float32 else 4e-2
    torch.testing.assert_close(buf.cpu().float(), ref.float(),
                               atol=tol, rtol=tol)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_weighted_delta_accum(dtype):
    from olearning_sim_amd.ops import fused
    C = 7
    buf, grad, master, offs = _layout(SHAPES, C, dtype)
    w = torch.rand(C).cuda()
    P = master.numel()
    delta = torch.randn(P).cuda()
    # fp32 torch reference using the same layout
    ref = delta.clone()
    offl = offs.tolist()
    for b in range(len(offl) - 1):
        g0, g1 = offl[b], offl[b + 1]
        n = g1 - g0
        blk = buf[C * g0:C * g1].view(C, n).float()
        ref[g0:g1] += ((blk - master[g0:g1].float().unsqueeze(0))
                       * w.unsqueeze(1)).sum(0)
    fused.weighted_delta_accum_flat(delta, buf, master, w, C, offsets=offs,
                                    wsum=float(w.sum()))
    tol = 1e-4 if dtype == torch.float32 else 5e-2
    torch.testing.assert_close(delta, ref, atol=tol, rtol=tol)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("K", [10, 100, 768, 30522])
def test_cross_entropy_matches_torch(dtype, K):
    from olearning_sim_amd.ops import fused
    N = 256
    g = torch.Generator().manual_seed(1)
    logits = (torch.randn(N, K, generator=g) * 3).to(dtype).cuda()
    labels = torch.randint(0, K, (N,), generator=g).cuda()
    lg = logits.detach().clone().requires_grad_(True)
    loss = fused.cross_entropy_fwd_bwd(lg, labels)
    loss.backward()
    ref_in = logits.detach().float().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(ref_in, labels)
    ref.backward()
    atol = 2e-3 if dtype == torch.float32 else 2e-2
    assert abs(float(loss) - float(ref)) < atol * max(1.0, abs(float(ref)))
    torch.testing.assert_close(lg.grad.float(), ref_in.grad,
                               atol=5e-3 if dtype == torch.float32 else 3e-2,
                               rtol=1e-2)


def test_engine_round_gpu_runs_and_trains():
    """Full engine rounds on GPU bf16: finite master, loss decreases."""
    from olearning_sim_amd.engine import EngineJob, LogicalEngine
    rows = []
    job = EngineJob(task_id="t", model_name="mlp",
                    model_kwargs={"in_features": 64, "hidden": 32,
                                  "num_classes": 10},
                    clients=16, rounds=6, local_steps=2, batch_size=8,
                    lr=0.1, device="cuda:0", dtype="bfloat16",
                    num_classes=10, seed=5)
    eng = LogicalEngine(job, result_sink=rows.append)
    out = eng.run()
    assert out["rounds"] == 6
    assert eng.master.flat.isfinite().all()
    losses = [r["loss"] for r in rows if r["loss"] is not None]
    assert losses[-1] < losses[0]


def test_gpu_engine_uses_hip_ops(monkeypatch):
    """The GPU path must go through the extension (no silent fallback)."""
    import olearning_sim_amd.ops.fused as F
    calls = {"n": 0}
    real = F.load_hip_ops

    def counting(required=False):
        calls["n"] += 1
        return real(required=required)

    monkeypatch.setattr(F, "load_hip_ops", counting)
    from olearning_sim_amd.engine import EngineJob, LogicalEngine
    job = EngineJob(task_id="t", model_name="mlp",
                    model_kwargs={"in_features": 32, "hidden": 16,
                                  "num_classes": 4},
                    clients=4, rounds=1, local_steps=1, batch_size=4,
                    lr=0.1, device="cuda:0", dtype="bfloat16", num_classes=4)
    LogicalEngine(job).run_round(0)
    assert calls["n"] > 0


@pytest.mark.parametrize("shape", [(4, 5, 64, 8, 8), (3, 7, 128, 16, 16)])
@pytest.mark.parametrize("relu,res", [(False, False), (True, False), (True, True)])
def test_groupnorm_fused_matches_reference(shape, relu, res):
    from olearning_sim_amd.ops.fused import groupnorm_act
    from olearning_sim_amd.models.base import bgroupnorm
    B, C, ch, H, W = shape
    G = 8
    g = torch.Generator().manual_seed(0)
    x = torch.randn(B, C * ch, H, W, generator=g).cuda().to(torch.bfloat16)
    gamma = (1 + 0.1 * torch.randn(C, ch, generator=g)).cuda().to(torch.bfloat16)
    beta = (0.1 * torch.randn(C, ch, generator=g)).cuda().to(torch.bfloat16)
    resid = (torch.randn(B, C * ch, H, W, generator=g).cuda().to(torch.bfloat16)
             if res else None)

    xf = x.detach().clone().requires_grad_(True)
    gf = gamma.detach().clone().requires_grad_(True)
    bf = beta.detach().clone().requires_grad_(True)
    rf = resid.detach().clone().requires_grad_(True) if res else None
    y = groupnorm_act(xf, C, G, gf, bf, res=rf, relu=relu)

    # fp32 torch reference
    xr = x.detach().float().requires_grad_(True)
    gr = gamma.detach().float().requires_grad_(True)
    br = beta.detach().float().requires_grad_(True)
    rr = resid.detach().float().requires_grad_(True) if res else None
    yr = bgroupnorm(xr, C, G, gr, br)
    if res:
        yr = yr + rr
    if relu:
        yr = torch.relu(yr)
    torch.testing.assert_close(y.float(), yr, atol=6e-2, rtol=6e-2)

    dy = torch.randn(y.shape, generator=g).cuda()
    y.backward(dy.to(y.dtype))
    yr.backward(dy)
    torch.testing.assert_close(xf.grad.float(), xr.grad, atol=1e-1, rtol=1e-1)
    torch.testing.assert_close(gf.grad.float(), gr.grad, atol=2e-1, rtol=5e-2)
    torch.testing.assert_close(bf.grad.float(), br.grad, atol=2e-1, rtol=5e-2)
    if res:
        torch.testing.assert_close(rf.grad.float(), rr.grad, atol=1e-1, rtol=1e-1)


def test_engine_per_tier_accounting_gpu():
    """Per-(data x tier) segment accounting with GPU behaviour sampling:
    vectors partition the cohort and feed the result rows."""
    import json as _json
    from olearning_sim_amd.engine import EngineJob, LogicalEngine
    strategy = _json.dumps({"offline_simulation": {
        "offline_probability": 0.5}})
    rows = []
    job = EngineJob(task_id="t_tier", model_name="mlp",
                    model_kwargs={"in_features": 64, "hidden": 32,
                                  "num_classes": 10},
                    clients=32, rounds=2, local_steps=1, batch_size=8,
                    lr=0.1, device="cuda:0", dtype="bfloat16",
                    num_classes=10, seed=9, dynamic_num=100,
                    behavior_strategy=strategy,
                    tier_counts=[("high", 20), ("low", 12)],
                    dynamic_nums=[100, 100])
    eng = LogicalEngine(job, result_sink=rows.append)
    out = eng.run()
    assert out["rounds"] == 2
    for rec in out["records"]:
        s, f = rec["success_per_tier"], rec["failed_per_tier"]
        assert s[0] + f[0] == 20 and s[1] + f[1] == 12
        assert rec["success"] == sum(s) and rec["failed"] == sum(f)
    tgt = rows[0]["logical_result"]["logical_result"][0]["simulation_target"]
    assert tgt["devices"] == ["high", "low"]
    assert eng.master.flat.isfinite().all()


@pytest.mark.parametrize("n", [4096, 36864, 100, 450, 10])
def test_weighted_delta_accum_single_param(n):
    """Single-param call path (nblocks==1): n%8==0 rides the vectorised
    k_delta_accum_v8 kernel, n%8!=0 the scalar fallback."""
    from olearning_sim_amd.ops import fused
    torch.manual_seed(3)
    C = 130
    buf = torch.randn(C * n, device="cuda").to(torch.bfloat16)
    master = torch.randn(n, device="cuda").to(torch.bfloat16)
    w = torch.rand(C, device="cuda")
    delta = torch.randn(n, device="cuda")
    offs = torch.tensor([0, n], dtype=torch.int64, device="cuda")
    ref = delta + ((buf.view(C, n).float()
                    - master.float().unsqueeze(0)) * w.unsqueeze(1)).sum(0)
    fused.weighted_delta_accum_flat(delta, buf, master, w, C, offsets=offs,
                                    wsum=float(w.sum()))
    torch.testing.assert_close(delta, ref, atol=5e-2, rtol=5e-2)


@pytest.mark.parametrize("shape", [(3, 64, 128), (2, 30522, 768),
                                   (5, 100, 70), (1, 64, 64)])
def test_transpose2d_matches_torch(shape):
    """LDS-tiled batched transpose == transpose().contiguous()."""
    from olearning_sim_amd.ops.fused import fast_transpose
    torch.manual_seed(5)
    for dt in (torch.bfloat16, torch.float32):
        x = torch.randn(*shape, device="cuda").to(dt)
        y = fast_transpose(x)
        ref = x.transpose(1, 2).contiguous()
        assert y.shape == ref.shape and y.is_contiguous()
        torch.testing.assert_close(y, ref, atol=0, rtol=0)


@pytest.mark.parametrize("shape", [(5, 64, 16, 32, 32), (2, 3, 4, 8, 8),
                                   (7, 16, 16)])
def test_subsample2_matches_slicing(shape):
    """subsample2 fwd/bwd == x[..., ::2, ::2] slicing autograd."""
    from olearning_sim_amd.ops.conv import _Subsample2Fn
    torch.manual_seed(6)
    for dt in (torch.bfloat16, torch.float32):
        x = torch.randn(*shape, device="cuda").to(dt).requires_grad_(True)
        xr = x.detach().clone().requires_grad_(True)
        y = _Subsample2Fn.apply(x)
        ref = xr[..., ::2, ::2]
        torch.testing.assert_close(y, ref, atol=0, rtol=0)
        dy = torch.randn_like(y)
        y.backward(dy)
        ref.backward(dy)
        torch.testing.assert_close(x.grad, xr.grad, atol=0, rtol=0)


@pytest.mark.parametrize("shape", [(64,), (512, 512, 3, 3), (100,), (16, 8)])
def test_replicate_params_matches_clone(shape):
    """Broadcast-replicate kernel == expand().clone() (bitwise); shapes
    with numel % 8 != 0 take the torch fallback inside replicate_params."""
    from olearning_sim_amd.engine.client_manager import replicate_params
    torch.manual_seed(3)
    for dt in (torch.bfloat16, torch.float32):
        src = torch.randn(*shape, device="cuda").to(dt)
        out = replicate_params({"p": src}, 7)["p"]
        ref = src.unsqueeze(0).expand(7, *shape).contiguous()
        assert out.requires_grad and out.data_ptr() != src.data_ptr()
        torch.testing.assert_close(out.detach(), ref, atol=0, rtol=0)


def test_synth_batch_matches_composed():
    """Fused data-gen broadcast+shift+cast == the composed torch form."""
    from olearning_sim_amd.ops.fused import load_hip_ops
    ops = load_hip_ops(required=True)
    torch.manual_seed(4)
    B, C, n, K = 16, 25, 3072, 100
    x = torch.randn(B, n, device="cuda")
    y = torch.randint(0, K, (C, B), device="cuda")
    out = ops.synth_batch(x, y.reshape(-1).contiguous(), 0.1 / K, -0.05)
    ref = (x.unsqueeze(0) + 0.1 * (y.float().unsqueeze(2) / K - 0.5)) \
        .to(torch.bfloat16)
    torch.testing.assert_close(out.view(C, B, n), ref)


def test_relu_mask_matches_composed():
    """One-pass dy*(y>0) == the composed compare+cast+multiply chain."""
    from olearning_sim_amd.ops.fused import load_hip_ops
    ops = load_hip_ops(required=True)
    torch.manual_seed(12)
    for dt in (torch.bfloat16, torch.float32):
        y = torch.randn(40, 16, 24, device="cuda").to(dt)
        dy = torch.randn_like(y)
        out = ops.relu_mask(dy, y)
        ref = dy * (y > 0).to(dt)
        torch.testing.assert_close(out, ref, atol=0, rtol=0)


@pytest.mark.parametrize("shape,pad", [
    ((250, 16, 10, 32, 32), 1), ((250, 64, 10, 8, 8), 1),
    ((100, 6, 20, 24, 24), 4), ((7, 3, 5, 14, 18), 1),
])
def test_pad2d_matches_fpad(shape, pad):
    """Single-pass pad kernel == F.pad (bitwise, bf16)."""
    import torch.nn.functional as F
    from olearning_sim_amd.ops.conv import _pad
    torch.manual_seed(11)
    x = torch.randn(*shape, device="cuda").to(torch.bfloat16)
    y = _pad(x, pad)
    ref = F.pad(x, (pad, pad, pad, pad))
    torch.testing.assert_close(y, ref, atol=0, rtol=0)


@pytest.mark.parametrize("shape", [(3, 8, 768), (2, 512, 768), (5, 4, 128)])
def test_layernorm_matches_torch(shape):
    """Fused per-client LayerNorm fwd/bwd vs composed fp32 torch."""
    from olearning_sim_amd.ops.fused import layernorm
    torch.manual_seed(9)
    C, N, H = shape
    x0 = torch.randn(C, N, H) * 2 + 0.3
    g0 = torch.randn(C, H) * 0.5 + 1.0
    b0 = torch.randn(C, H) * 0.2
    dy = torch.randn(C, N, H) * 0.1

    xg = x0.to(torch.bfloat16).cuda().requires_grad_(True)
    gg = g0.to(torch.bfloat16).cuda().requires_grad_(True)
    bg = b0.to(torch.bfloat16).cuda().requires_grad_(True)
    y = layernorm(xg, gg, bg)
    assert y is not None
    y.backward(dy.to(torch.bfloat16).cuda())

    xr = x0.clone().requires_grad_(True)
    gr = g0.clone().requires_grad_(True)
    br = b0.clone().requires_grad_(True)
    mean = xr.mean(dim=-1, keepdim=True)
    var = xr.var(dim=-1, unbiased=False, keepdim=True)
    yr = (xr - mean) * torch.rsqrt(var + 1e-5) * gr.unsqueeze(1) \
        + br.unsqueeze(1)
    yr.backward(dy)

    torch.testing.assert_close(y.float().cpu(), yr, atol=5e-2, rtol=5e-2)
    torch.testing.assert_close(xg.grad.float().cpu(), xr.grad,
                               atol=5e-2, rtol=8e-2)
    torch.testing.assert_close(gg.grad.float().cpu(), gr.grad,
                               atol=0.3, rtol=5e-2)
    torch.testing.assert_close(bg.grad.float().cpu(), br.grad,
                               atol=0.3, rtol=5e-2)
