"""Micro-benchmark: BN backward two-pass vs one-pass (DDPX_BN1PASS)."""
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import torch

from ddp_tricks_amd.ops import load_extension

ext = load_extension(required=True)
CL = torch.channels_last
DEV = torch.device("cuda:0")

def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

for (N, C, H, W) in [(1024, 64, 56, 56), (1024, 256, 56, 56),
                     (1024, 512, 28, 28), (1024, 1024, 14, 14),
                     (1024, 2048, 7, 7)]:
    x = torch.randn(N, C, H, W, device=DEV).to(torch.bfloat16).contiguous(memory_format=CL)
    g = torch.randn(C, device=DEV).abs() + 0.5
    b = torch.randn(C, device=DEV)
    rm = torch.zeros(C, device=DEV)
    rv = torch.ones(C, device=DEV)
    y, sm, si, mask = ext.bn_fwd_train(x, g, b, rm, rv, 0.1, 1e-5, True)
    dy = torch.randn_like(x)
    os.environ["DDPX_BN1PASS"] = "0"
    t2 = bench(lambda: ext.bn_bwd(x, dy, g, sm, si, mask, True))
    os.environ["DDPX_BN1PASS"] = "1"
    t1 = bench(lambda: ext.bn_bwd(x, dy, g, sm, si, mask, True))
    os.environ["DDPX_BN1PASS"] = "0"
    mb = N * C * H * W * 2 * 2 / 1e6
    print(f"N{N} C{C} {H}x{W} ({mb:.0f} MB x+dy): 2pass {t2:8.1f} us   "
          f"1pass {t1:8.1f} us   speedup {t2 / t1:.2f}x", flush=True)
