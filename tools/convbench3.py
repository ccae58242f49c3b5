"""Per-shape conv A/B: hand-written MFMA kernels vs MIOpen grouped conv.

Measures fwd / dgrad / wgrad separately for every ResNet-18 CIFAR shape
(the headline bench's conv work) at a given client count, printing ms
and TF/s per direction plus the round-trip totals.  MIOpen backward is
split with aten.convolution_backward's output_mask.

Usage (on the GPU box):
    python tools/convbench3.py [--clients 250] [--iters 20] [--check]
"""

import argparse
import os
import shutil
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def _seed_miopen_find_db() -> None:
    if os.environ.get("MIOPEN_USER_DB_PATH"):
        return
    src = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "olearning_sim_amd", "ops", "miopen_udb")
    if not os.path.isdir(src):
        return
    dst = os.path.join(tempfile.gettempdir(),
                       f"olsim_miopen_udb_{os.getuid()}")
    os.makedirs(dst, exist_ok=True)
    for f in os.listdir(src):
        target = os.path.join(dst, f)
        if not os.path.exists(target):
            tmp = target + f".tmp{os.getpid()}"
            shutil.copy2(os.path.join(src, f), tmp)
            os.replace(tmp, target)
    os.environ["MIOPEN_USER_DB_PATH"] = dst


_seed_miopen_find_db()

import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402

from olearning_sim_amd.ops import load_hip_ops  # noqa: E402

# (name, ic, oc, h, stride) — every distinct 3x3 conv of ResNet-18 CIFAR
SHAPES = [
    ("stem", 3, 64, 32, 1),
    ("s0", 64, 64, 32, 1),
    ("s1d", 64, 128, 32, 2),
    ("s1", 128, 128, 16, 1),
    ("s2d", 128, 256, 16, 2),
    ("s2", 256, 256, 8, 1),
    ("s3d", 256, 512, 8, 2),
    ("s3", 512, 512, 4, 1),
]


def timeit(fn, iters, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--clients", type=int, default=250)
    ap.add_argument("--batch", type=int, default=16)
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--check", action="store_true",
                    help="numerics check vs fp32 reference at small C")
    ap.add_argument("--shapes", type=str, default="",
                    help="comma-separated subset of shape names")
    args = ap.parse_args()

    ops = load_hip_ops(required=True)
    C, B = args.clients, args.batch
    dev, dt = "cuda", torch.bfloat16
    names = set(args.shapes.split(",")) if args.shapes else None

    tot = {"cust": 0.0, "mio": 0.0, "v6": 0.0}
    print(f"# C={C} B={B} iters={args.iters} "
          f"(ms per call; TF = 2*C*B*OC*IC*9*OH*OW / ms)")
    print(f"{'shape':>5} {'dir':>5} | {'custom ms':>9} {'TF':>6} | "
          f"{'miopen ms':>9} {'TF':>6} | ratio")
    for name, ic, oc, h, st in SHAPES:
        if names and name not in names:
            continue
        oh = h // st
        x = torch.randn(C, ic, B, h, h, device=dev, dtype=dt)
        w = (torch.randn(C, oc, ic, 3, 3, device=dev, dtype=dt) * 0.05)
        y = ops.conv3x3_fwd(x, w, st)
        dy = torch.randn_like(y)
        # channel-grouped layout for MIOpen
        xg = x.permute(2, 0, 1, 3, 4).reshape(B, C * ic, h, h).contiguous()
        wf = w.reshape(C * oc, ic, 3, 3).contiguous()
        yg = F.conv2d(xg, wf, stride=st, padding=1, groups=C)
        dyg = dy.permute(2, 0, 1, 3, 4).reshape(B, C * oc, oh, oh).contiguous()

        flops = 2.0 * C * B * oc * ic * 9 * oh * oh

        def mio_bwd(mask):
            return torch.ops.aten.convolution_backward(
                dyg, xg, wf, None, [st, st], [1, 1], [1, 1], False,
                [0, 0], C, mask)

        v6 = bool(ops.conv3x3_v6_ok(ic, oc, B, h, h, st))
        x_pad = F.pad(x, (1, 1, 1, 1)) if v6 else None

        rows = [
            ("fwd", lambda: ops.conv3x3_fwd(x, w, st),
             lambda: F.conv2d(xg, wf, stride=st, padding=1, groups=C),
             (lambda: ops.conv3x3_fwd_p(F.pad(x, (1, 1, 1, 1)), w, st))
             if v6 else None),
            ("dgrad", lambda: ops.conv3x3_dgrad(dy, w, h, h, st),
             lambda: mio_bwd([True, False, False]),
             (lambda: ops.conv3x3_dgrad_p(F.pad(dy, (1, 1, 1, 1)), w, h, h,
                                          st)) if v6 else None),
            ("wgrad", lambda: ops.conv3x3_wgrad(x, dy, st),
             lambda: mio_bwd([False, True, False]),
             (lambda: ops.conv3x3_wgrad_p(x_pad, dy, st)) if v6 else None),
        ]
        for dname, cfn, mfn, v6fn in rows:
            cms = timeit(cfn, args.iters)
            mms = timeit(mfn, args.iters)
            v6ms = timeit(v6fn, args.iters) if v6fn else float("nan")
            tot["cust"] += cms
            tot["mio"] += mms
            tot["v6"] += v6ms if v6fn else cms
            print(f"{name:>5} {dname:>5} | {cms:9.3f} {flops/cms/1e9:6.0f} | "
                  f"{mms:9.3f} {flops/mms/1e9:6.0f} | {cms/mms:5.2f} | "
                  f"v6 {v6ms:8.3f} {flops/v6ms/1e9 if v6fn else 0:6.0f} "
                  f"{v6ms/mms:5.2f}",
                  flush=True)

        if args.check:
            xr = x.float()
            wr = w.float()
            xgr = xr.permute(2, 0, 1, 3, 4).reshape(B, C * ic, h, h)
            xgr.requires_grad_(True)
            wgr = wr.reshape(C * oc, ic, 3, 3).detach().requires_grad_(True)
            ygr = F.conv2d(xgr, wgr, stride=st, padding=1, groups=C)
            ygr.backward(dyg.float())
            yref = ygr.detach().reshape(B, C, oc, oh, oh).permute(1, 2, 0, 3, 4)
            dxref = xgr.grad.reshape(B, C, ic, h, h).permute(1, 2, 0, 3, 4)
            dwref = wgr.grad.reshape(C, oc, ic, 3, 3)
            err = (y.float() - yref).abs().max().item()
            scale = yref.abs().max().item()
            print(f"  {name} fwd maxerr {err:.4f} (|y|max {scale:.2f})")
            if v6:
                y6 = ops.conv3x3_fwd_p(x_pad, w, st)
                e6 = (y6.float() - yref).abs().max().item()
                dx6 = ops.conv3x3_dgrad_p(F.pad(dy, (1, 1, 1, 1)), w, h, h, st)
                edx = (dx6.float() - dxref).abs().max().item()
                sdx = dxref.abs().max().item()
                dw6 = ops.conv3x3_wgrad_p(x_pad, dy, st)
                edw = (dw6.float() - dwref).abs().max().item()
                sdw = dwref.abs().max().item()
                print(f"  {name} v6 fwd {e6:.4f} dgrad {edx:.4f} "
                      f"(|dx|max {sdx:.2f}) wgrad {edw:.4f} "
                      f"(|dw|max {sdw:.2f})")
        del x, w, y, dy, xg, wf, yg, dyg, x_pad
        torch.cuda.empty_cache()

    print(f"\nTOTAL custom {tot['cust']:.2f} ms  vs  miopen {tot['mio']:.2f} "
          f"ms   ratio {tot['cust']/max(1e-9, tot['mio']):.3f}   "
          f"v6 {tot['v6']:.2f} ms ratio {tot['v6']/max(1e-9, tot['mio']):.3f}")


if __name__ == "__main__":
    main()
