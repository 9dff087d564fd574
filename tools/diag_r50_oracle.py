"""Diagnose the resnet50 GPU-vs-CPU oracle: worst grad cosines with
skip-fusion on/off, plus grad norms (is the outlier near-zero noise?)."""
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import torch

from ddp_tricks_amd import amp, same_seeds
from ddp_tricks_amd.models import build_model
from ddp_tricks_amd.ops.functional import clear_weight_cache, cross_entropy_loss

DEV = torch.device("cuda:0")


def run(dev, arch="resnet50"):
    amp._state.__init__()
    clear_weight_cache()
    same_seeds(3)
    model = build_model(arch, num_classes=10, cifar_stem=True).to(dev)
    if dev.type == "cuda":
        model, _ = amp.initialize(model, None, opt_level="O1")
    g = torch.Generator().manual_seed(4)
    x = torch.rand(16, 3, 32, 32, generator=g).to(dev)
    t = torch.randint(0, 10, (16,), generator=g).to(dev)
    model.train()
    out = model(x)
    loss = cross_entropy_loss(out, t)
    loss.backward()
    grads = {k: p.grad.detach().float().cpu()
             for k, p in model.named_parameters()}
    amp._state.__init__()
    return grads


cpu = run(torch.device("cpu"))
for sf in ("1", "0"):
    os.environ["DDPX_SKIPFUSE"] = sf
    gpu = run(DEV)
    coses = {}
    for k in cpu:
        coses[k] = torch.nn.functional.cosine_similarity(
            gpu[k].flatten(), cpu[k].flatten(), dim=0).item()
    worst = sorted(coses.items(), key=lambda kv: kv[1])[:8]
    print(f"--- DDPX_SKIPFUSE={sf}: worst cosines ---")
    for k, c in worst:
        print(f"  {k:34s} cos={c:+.4f}  |g_cpu|={cpu[k].norm():.3e} "
              f"|g_gpu|={gpu[k].norm():.3e}")
