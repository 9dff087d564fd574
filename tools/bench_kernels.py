"""Micro-benchmark of the HIP conv kernels at the workload's shapes.

Usage (on a GPU box):  python tools/bench_kernels.py [--resnet]

Times conv fwd/dgrad/wgrad per shape with HIP events (200 reps after 20
warmup) and prints achieved TFLOP/s next to microseconds, so kernel A/B
decisions are measurements, not guesses.  DDPX_WGRAD_V=sb selects the
single-buffer wgrad variant for comparison.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from ddp_tricks_amd.ops import load_extension  # noqa: E402

CL = torch.channels_last

TOYNET = [
    # (name, N, C, H, W, K, R, stride, pad)
    ("conv2", 1024, 64, 26, 26, 128, 3, 1, 0),
    ("conv3", 1024, 128, 12, 12, 256, 3, 1, 0),
    ("conv4", 1024, 256, 10, 10, 512, 3, 1, 0),
    ("conv1", 1024, 1, 28, 28, 64, 3, 1, 0),
    ("conv1p8", 1024, 8, 28, 28, 64, 3, 1, 0),
]
RESNET = [
    ("r18s1", 256, 64, 32, 32, 64, 3, 1, 1),
    ("r18s2", 256, 64, 32, 32, 128, 3, 2, 1),
    ("r18d2", 256, 64, 32, 32, 128, 1, 2, 0),
    ("r50c1", 256, 64, 56, 56, 64, 1, 1, 0),
    ("r50c3", 256, 64, 56, 56, 256, 1, 1, 0),
    ("r50l3", 256, 256, 14, 14, 1024, 1, 1, 0),
    ("stem50", 64, 8, 224, 224, 64, 7, 2, 3),
]


def time_op(fn, reps=200, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(reps):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / reps * 1000.0  # us


def bench_functional(shapes, reps):
    """Times the autograd-level conv path (true dispatch: 1x1 GEMM route,
    pad8 stems, fused wgrad variants) — fwd and full bwd per shape."""
    import torch.nn as nn

    from ddp_tricks_amd.ops import functional as F_ops
    dev = torch.device("cuda:0")
    print(f"{'shape':8s} {'op':6s} {'us':>9s} {'TFLOP/s':>9s}")
    for name, N, C, H, W, K, R, stride, pad in shapes:
        P = (H + 2 * pad - R) // stride + 1
        x = torch.randn(N, C, H, W, device=dev).to(torch.bfloat16)\
            .contiguous(memory_format=CL).requires_grad_(True)
        w = nn.Parameter(torch.randn(K, C, R, R, device=dev))
        dy = torch.randn(N, K, P, P, device=dev).to(torch.bfloat16)\
            .contiguous(memory_format=CL)
        flops = 2.0 * N * P * P * K * C * R * R

        us = time_op(lambda: F_ops.conv2d(x.detach(), w, None, stride, pad),
                     reps)
        print(f"{name:8s} {'Ffwd':6s} {us:9.1f} {flops/us/1e6:9.1f}")

        def fb():
            y = F_ops.conv2d(x, w, None, stride, pad)
            y.backward(dy)
        us = time_op(fb, reps // 2)
        print(f"{name:8s} {'Ffb':6s} {us:9.1f} {3*flops/us/1e6:9.1f}")


BN_SHAPES = [
    ("stem", 256, 64, 112, 112),
    ("l1", 256, 256, 56, 56),
    ("l1b", 256, 64, 56, 56),
    ("l2", 256, 512, 28, 28),
    ("l3", 256, 1024, 14, 14),
    ("l4", 256, 2048, 7, 7),
    ("toy2", 1024, 128, 24, 24),
]


def bench_bn(reps):
    ext = load_extension(required=True)
    dev = torch.device("cuda:0")
    print(f"{'shape':6s} {'op':10s} {'us':>9s} {'GB/s':>8s}")
    for name, N, C, H, W in BN_SHAPES:
        x = torch.randn(N, C, H, W, device=dev).to(torch.bfloat16).contiguous(memory_format=CL)
        dy = torch.randn_like(x).contiguous(memory_format=CL)
        g = torch.rand(C, device=dev) + 0.5
        b = torch.randn(C, device=dev)
        rm = torch.zeros(C, device=dev)
        rv = torch.ones(C, device=dev)
        nbytes = x.numel() * 2
        us = time_op(lambda: ext.bn_fwd_train(x, g, b, rm, rv, 0.1, 1e-5,
                                              True, None), reps)
        # fwd: read x twice (partial+apply) + write y + mask
        print(f"{name:6s} {'fwd_train':10s} {us:9.1f} {(3.06*nbytes)/us/1e3:8.0f}")
        y, sm, si, mask = ext.bn_fwd_train(x, g, b, rm, rv, 0.1, 1e-5, True, None)
        us = time_op(lambda: ext.bn_bwd(x, dy, g, sm, si, mask, True, False),
                     reps)
        # bwd: partial reads x,dy,mask; dx reads x,dy,mask writes dx
        print(f"{name:6s} {'bwd':10s} {us:9.1f} {(5.13*nbytes)/us/1e3:8.0f}")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--resnet", action="store_true")
    ap.add_argument("--functional", action="store_true")
    ap.add_argument("--bn", action="store_true")
    ap.add_argument("--reps", type=int, default=200)
    args = ap.parse_args()
    if args.bn:
        bench_bn(args.reps)
        return
    if args.functional:
        load_extension(required=True)
        bench_functional(TOYNET + (RESNET if args.resnet else []), args.reps)
        return
    ext = load_extension(required=True)
    dev = torch.device("cuda:0")
    shapes = TOYNET + (RESNET if args.resnet else [])
    print(f"{'shape':8s} {'op':6s} {'us':>9s} {'TFLOP/s':>9s}")
    for name, N, C, H, W, K, R, stride, pad in shapes:
        P = (H + 2 * pad - R) // stride + 1
        x = torch.randn(N, C, H, W, device=dev).to(torch.bfloat16).contiguous(memory_format=CL)
        w = torch.randn(K, C, R, R, device=dev).to(torch.bfloat16).contiguous(memory_format=CL)
        dy = torch.randn(N, K, P, P, device=dev).to(torch.bfloat16).contiguous(memory_format=CL)
        wt2 = w.to(torch.bfloat16).permute(1, 2, 3, 0).reshape(C, R * R * K).contiguous()
        flops = 2.0 * N * P * P * K * C * R * R
        us = time_op(lambda: ext.conv2d_fwd(x, w, None, stride, pad), args.reps)
        print(f"{name:8s} {'fwd':6s} {us:9.1f} {flops/us/1e6:9.1f}")
        if C >= 8:
            us = time_op(lambda: ext.conv2d_dgrad(dy, wt2, N, C, H, W, R, R,
                                                  stride, pad), args.reps)
            print(f"{name:8s} {'dgrad':6s} {us:9.1f} {flops/us/1e6:9.1f}")
        us = time_op(lambda: ext.conv2d_wgrad(dy, x, R, R, stride, pad),
                     args.reps)
        print(f"{name:8s} {'wgrad':6s} {us:9.1f} {flops/us/1e6:9.1f}")


if __name__ == "__main__":
    main()
