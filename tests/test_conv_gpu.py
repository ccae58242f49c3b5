"""Numerics of the hand-written client-batched conv3x3 MFMA kernels
against the fp32 torch grouped-conv reference."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _require_ext():
    from olearning_sim_amd.ops import load_hip_ops
    load_hip_ops(required=True)


def test_mfma_fragment_layout_selftest():
    """16x16x32 bf16 MFMA fragment maps vs torch mm."""
    from olearning_sim_amd.ops.fused import load_hip_ops
    ops = load_hip_ops(required=True)
    g = torch.Generator().manual_seed(0)
    A = torch.randn(16, 32, generator=g).to(torch.bfloat16).cuda()
    B = torch.randn(32, 16, generator=g).to(torch.bfloat16).cuda()
    D = ops.mfma_selftest(A, B)
    ref = (A.float() @ B.float())
    torch.testing.assert_close(D, ref, atol=5e-2, rtol=5e-2)


SHAPES = [
    # (C, IC, OC, B, H, stride)
    (3, 64, 64, 4, 16, 1),
    (2, 64, 128, 4, 16, 2),
    (2, 16, 16, 2, 8, 1),
    (2, 3, 64, 4, 32, 1),      # stem: IC=3 (K=27, sub-tile)
    (2, 512, 512, 2, 4, 1),    # deepest stage
    (1, 128, 256, 3, 8, 2),
    # v6 template variants (client_conv2.hip): stride-2 parity classes
    # at LG_OW 4/3/2 and small-plane fwd/dgrad sub-row hoisting
    (2, 64, 128, 4, 32, 2),    # s2 classes, dy plane 16x16 (LG4)
    (2, 128, 256, 4, 16, 2),   # s2 classes, dy plane 8x8 (LG3)
    (2, 256, 512, 2, 8, 2),    # s2 classes, dy plane 4x4 (LG2)
    (2, 64, 64, 4, 8, 1),      # fwd/dgrad LG3 sub-rows
    (2, 128, 128, 2, 4, 1),    # fwd/dgrad LG2 sub-rows
    (9, 64, 64, 2, 16, 1),     # C not a multiple of 8: XCD remap tail
]


def test_v6_matches_v5_paths():
    """The padded v6 kernels must agree with the v5 family on the same
    inputs (both bf16; tolerance covers bf16 accumulation-order drift)."""
    import os
    from olearning_sim_amd.ops.conv import client_conv3x3
    g = torch.Generator().manual_seed(7)
    for (C, IC, OC, B, H, st) in [(3, 64, 64, 4, 16, 1),
                                  (2, 64, 128, 4, 16, 2),
                                  (2, 128, 128, 2, 8, 1)]:
        x0 = torch.randn(C, IC, B, H, H, generator=g) * 0.5
        w0 = torch.randn(C, OC, IC, 3, 3, generator=g) * 0.1
        dy0 = torch.randn(C, OC, B, H // st, H // st, generator=g) * 0.1
        outs = {}
        for ver in ("5", "6"):
            os.environ["OLSIM_CONV_V"] = ver if ver == "5" else ""
            try:
                xg = x0.to(torch.bfloat16).cuda().requires_grad_(True)
                wg = w0.to(torch.bfloat16).cuda().requires_grad_(True)
                y = client_conv3x3(xg, wg, st)
                y.backward(dy0.to(torch.bfloat16).cuda())
                outs[ver] = (y.detach(), xg.grad.clone(), wg.grad.clone())
            finally:
                os.environ.pop("OLSIM_CONV_V", None)
        for a, b in zip(outs["5"], outs["6"]):
            torch.testing.assert_close(a.float(), b.float(),
                                       atol=5e-2, rtol=5e-2)


@pytest.mark.parametrize("C,IC,OC,B,H,stride", SHAPES)
def test_conv3x3_fwd_matches_reference(C, IC, OC, B, H, stride):
    from olearning_sim_amd.ops.conv import client_conv3x3, _cpu_conv3x3
    g = torch.Generator().manual_seed(1)
    x = torch.randn(C, IC, B, H, H, generator=g).to(torch.bfloat16).cuda()
    w = (torch.randn(C, OC, IC, 3, 3, generator=g) * 0.1).to(torch.bfloat16).cuda()
    y = client_conv3x3(x, w, stride)
    ref = _cpu_conv3x3(x.float().cpu(), w.float().cpu(), stride)
    torch.testing.assert_close(y.float().cpu(), ref,
                               atol=0.1 * (IC ** 0.5) * 0.3, rtol=5e-2)


@pytest.mark.parametrize("C,IC,OC,B,H,stride", SHAPES)
def test_conv3x3_backward_matches_reference(C, IC, OC, B, H, stride):
    from olearning_sim_amd.ops.conv import client_conv3x3, _cpu_conv3x3
    g = torch.Generator().manual_seed(2)
    x0 = torch.randn(C, IC, B, H, H, generator=g) * 0.5
    w0 = torch.randn(C, OC, IC, 3, 3, generator=g) * 0.1

    xg = x0.to(torch.bfloat16).cuda().requires_grad_(True)
    wg = w0.to(torch.bfloat16).cuda().requires_grad_(True)
    y = client_conv3x3(xg, wg, stride)
    dy = torch.randn(y.shape, generator=g) * 0.1
    y.backward(dy.to(torch.bfloat16).cuda())

    xr = x0.clone().requires_grad_(True)
    wr = w0.clone().requires_grad_(True)
    yr = _cpu_conv3x3(xr, wr, stride)
    yr.backward(dy)

    scale = max(1.0, float(yr.abs().max()))
    torch.testing.assert_close(xg.grad.float().cpu(), xr.grad,
                               atol=5e-2 * (OC ** 0.5) * 0.1 + 2e-2, rtol=8e-2)
    torch.testing.assert_close(wg.grad.float().cpu(), wr.grad,
                               atol=2e-2 * (B * H * H) ** 0.5 * 0.1 + 2e-2,
                               rtol=8e-2)


def test_conv1x1_matches_reference():
    from olearning_sim_amd.ops.conv import client_conv1x1
    g = torch.Generator().manual_seed(3)
    C, IC, OC, B, H = 3, 64, 128, 4, 16
    for stride in (1, 2):
        x = torch.randn(C, IC, B, H, H, generator=g).to(torch.bfloat16).cuda()
        w = (torch.randn(C, OC, IC, 1, 1, generator=g) * 0.1).to(torch.bfloat16).cuda()
        y = client_conv1x1(x, w, stride)
        xs = x.float()[:, :, :, ::stride, ::stride]
        ref = torch.einsum("coi,cibhw->cobhw", w.float().squeeze(-1).squeeze(-1), xs)
        torch.testing.assert_close(y.float(), ref.cuda(), atol=5e-1, rtol=5e-2)


def test_resnet_cbf_forward_matches_grouped_path():
    """The MFMA fast path must equal the grouped-conv reference path."""
    from olearning_sim_amd.models import build_model
    from olearning_sim_amd.engine.client_manager import (FlatParams,
                                                         replicate_params)
    m = build_model("resnet18", num_classes=10, width_mult=0.25)
    gen = torch.Generator().manual_seed(0)
    gp = m.init_global(generator=gen)
    gp = {k: v.cuda() for k, v in gp.items()}
    master = FlatParams(gp)
    C, B = 3, 2
    params = replicate_params(master.cast(torch.bfloat16), C)
    x = torch.randn(C, B, 3, 32, 32, generator=gen).to(torch.bfloat16).cuda()
    fast = m.forward_cbf(params, x)
    # reference grouped path with the same bf16 params
    C2, B2 = x.shape[0], x.shape[1]
    import torch.nn.functional as F
    ref = m.forward.__wrapped__ if hasattr(m.forward, "__wrapped__") else None
    # run the grouped path by forcing non-cbf branch: use fp32 CPU copy
    params32 = {k: v.detach().float().cpu() for k, v in params.items()}
    ref_out = m.forward(params32, x.float().cpu())
    torch.testing.assert_close(fast.float().cpu(), ref_out,
                               atol=0.25, rtol=0.1)


def test_resnet_cbf_bench_speed():
    """Sanity: fast path fwd+bwd runs and is not slower than 2x the
    grouped path (informational timing printed)."""
    import time
    from olearning_sim_amd.models import build_model
    from olearning_sim_amd.engine.client_manager import (FlatParams,
                                                         replicate_params)
    m = build_model("resnet18", num_classes=100)
    gen = torch.Generator().manual_seed(0)
    gp = {k: v.cuda() for k, v in m.init_global(generator=gen).items()}
    master = FlatParams(gp)
    C, B = 64, 16
    params = replicate_params(master.cast(torch.bfloat16), C)
    x = torch.randn(C, B, 3, 32, 32, generator=gen).to(torch.bfloat16).cuda()
    y = torch.randint(0, 100, (C, B), generator=gen).cuda()

    def step():
        loss = m.loss(params, x, y)
        torch.autograd.grad(loss, list(params.values()), allow_unused=True)

    step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(3):
        step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 3
    print(f"\ncbf fwd+bwd C={C} B={B}: {dt*1000:.1f} ms")
    assert dt < 10.0


CONV5_SHAPES = [
    # (C, IC, OC, B, H)
    (3, 3, 6, 8, 32),     # LeNet conv1
    (2, 6, 16, 8, 14),    # LeNet conv2
    (9, 3, 6, 16, 32),    # C not a multiple of 8 (XCD remap tail)
]


@pytest.mark.parametrize("C,IC,OC,B,H", CONV5_SHAPES)
@pytest.mark.parametrize("relu", [False, True])
def test_conv5x5_matches_reference(C, IC, OC, B, H, relu):
    """Valid 5x5 conv fwd+bwd (fused bias/ReLU) vs fp32 torch."""
    from olearning_sim_amd.ops.conv import client_conv5x5, _cpu_conv5x5
    g = torch.Generator().manual_seed(11)
    x0 = torch.randn(C, IC, B, H, H, generator=g) * 0.5
    w0 = torch.randn(C, OC, IC, 5, 5, generator=g) * 0.1
    b0 = torch.randn(C, OC, generator=g) * 0.1

    xg = x0.to(torch.bfloat16).cuda().requires_grad_(True)
    wg = w0.to(torch.bfloat16).cuda().requires_grad_(True)
    bg = b0.to(torch.bfloat16).cuda().requires_grad_(True)
    y = client_conv5x5(xg, wg, bg, relu=relu)
    dy = torch.randn(y.shape, generator=g) * 0.1
    y.backward(dy.to(torch.bfloat16).cuda())

    xr = x0.clone().requires_grad_(True)
    wr = w0.clone().requires_grad_(True)
    br = b0.clone().requires_grad_(True)
    yr = _cpu_conv5x5(xr, wr, br, relu)
    if relu:
        # bf16 y and fp32 y disagree on sign for pre-activations near 0,
        # flipping the ReLU mask at a few positions; compare gradients
        # under the KERNEL's mask (fwd output itself is checked below)
        dy_eff = dy * (y.float().cpu() > 0)
        yr_lin = _cpu_conv5x5(xr, wr, br, False)
        yr_lin.backward(dy_eff)
    else:
        yr.backward(dy)

    torch.testing.assert_close(y.float().cpu(), yr,
                               atol=0.05 * (IC ** 0.5) + 0.02, rtol=5e-2)
    torch.testing.assert_close(xg.grad.float().cpu(), xr.grad,
                               atol=0.05 * (OC ** 0.5) + 0.02, rtol=8e-2)
    torch.testing.assert_close(wg.grad.float().cpu(), wr.grad,
                               atol=2e-2 * (B * H * H) ** 0.5 * 0.1 + 2e-2,
                               rtol=8e-2)
    torch.testing.assert_close(bg.grad.float().cpu(), br.grad,
                               atol=0.5, rtol=5e-2)


def test_lenet_cbf_matches_grouped_path():
    """LeNet's conv5x5 MFMA fast path == grouped reference path."""
    from olearning_sim_amd.models import build_model
    from olearning_sim_amd.engine.client_manager import (FlatParams,
                                                         replicate_params)
    m = build_model("lenet")
    gen = torch.Generator().manual_seed(0)
    gp = {k: v.cuda() for k, v in m.init_global(generator=gen).items()}
    master = FlatParams(gp)
    C, B = 5, 8
    params = replicate_params(master.cast(torch.bfloat16), C)
    x = torch.randn(C, B, 3, 32, 32, generator=gen).to(torch.bfloat16).cuda()
    fast = m.forward_cbf(params, x)
    params32 = {k: v.detach().float().cpu() for k, v in params.items()}
    ref = m.forward(params32, x.float().cpu())
    torch.testing.assert_close(fast.float().cpu(), ref, atol=0.15, rtol=0.08)


def test_pool2x2_matches_torch():
    """Fused 2x2 max-pool fwd/bwd vs F.max_pool2d (argmax-routed grad)."""
    from olearning_sim_amd.ops.conv import max_pool2x2
    g = torch.Generator().manual_seed(21)
    for shape in [(5, 6, 8, 28, 28), (3, 16, 4, 10, 10), (2, 64, 64)]:
        x0 = torch.randn(*shape, generator=g)
        xg = x0.to(torch.bfloat16).cuda().requires_grad_(True)
        y = max_pool2x2(xg)
        dy = torch.randn(y.shape, generator=g)
        y.backward(dy.to(torch.bfloat16).cuda())

        xr = x0.to(torch.bfloat16).requires_grad_(True)
        flat = xr.reshape(-1, 1, shape[-2], shape[-1])
        yr = torch.nn.functional.max_pool2d(flat, 2).reshape(y.shape)
        yr.backward(dy.to(torch.bfloat16))
        torch.testing.assert_close(y.float().cpu(), yr.float(),
                                   atol=0, rtol=0)
        # grads may differ only where a 2x2 window has ties (argmax
        # routing order); values at the chosen positions are identical
        diff = (xg.grad.float().cpu() - xr.grad.float()).abs()
        assert float(diff.sum()) < 1e-3 or \
            float((diff > 0).float().mean()) < 0.01


@pytest.mark.parametrize("shape", [
    (3, 64, 64, 4, 16, 1), (2, 64, 32, 2, 8, 1), (2, 128, 64, 2, 32, 1),
    (1, 512, 512, 2, 8, 1),
    # stride 2 (v8s): OW 16 and 8
    (2, 64, 128, 4, 32, 2), (2, 128, 256, 2, 16, 2),
])
def test_wgrad_v8_bitwise_matches_scalar_gather(shape):
    """Run-vectorised wgrad producer (v8/v8s) == scalar-gather producer,
    bitwise (same MFMA tiles, same reduction order)."""
    import os
    import torch.nn.functional as F
    from olearning_sim_amd.ops.fused import load_hip_ops
    ops = load_hip_ops(required=True)
    C, IC, OC, B, H, stride = shape
    torch.manual_seed(9)
    OH = H // stride
    x = torch.randn(C, IC, B, H, H, device="cuda").to(torch.bfloat16)
    dy = torch.randn(C, OC, B, OH, OH, device="cuda").to(torch.bfloat16)
    xp = F.pad(x, (1, 1, 1, 1)).contiguous()
    assert "OLSIM_CONV_DW8" not in os.environ
    dw_v8 = ops.conv3x3_wgrad_p(xp, dy, stride)
    os.environ["OLSIM_CONV_DW8"] = "0"
    try:
        dw_v6 = ops.conv3x3_wgrad_p(xp, dy, stride)
    finally:
        del os.environ["OLSIM_CONV_DW8"]
    torch.testing.assert_close(dw_v8, dw_v6, atol=0, rtol=0)
