import os, sys
sys.path.insert(0, "/root/repo")
import torch
from ddp_tricks_amd import amp, same_seeds
from ddp_tricks_amd.models import build_model
from ddp_tricks_amd.ops.functional import clear_weight_cache, cross_entropy_loss

def run(dev):
    amp._state.__init__()
    clear_weight_cache()
    same_seeds(3)
    m = build_model("vgg16", num_classes=10, cifar_head=True).to(dev)
    if dev.type == "cuda":
        m, _ = amp.initialize(m, None, opt_level="O1")
    g = torch.Generator().manual_seed(4)
    x = torch.rand(16, 3, 32, 32, generator=g).to(dev)
    t = torch.randint(0, 10, (16,), generator=g).to(dev)
    m.train()
    out = m(x)
    loss = cross_entropy_loss(out, t)
    loss.backward()
    gr = {k: p.grad.detach().float().cpu() for k, p in m.named_parameters()}
    amp._state.__init__()
    return gr

gc = run(torch.device("cpu"))
gg = run(torch.device("cuda:0"))
coses = {k: torch.nn.functional.cosine_similarity(
    gg[k].flatten(), gc[k].flatten(), dim=0).item() for k in gc}
vals = sorted(coses.items(), key=lambda kv: kv[1])
print("worst 8:", [(k, round(v, 3)) for k, v in vals[:8]])
print("median:", round(sorted(coses.values())[len(coses)//2], 4))
print("classifier.6:", round(coses["classifier.6.weight"], 4))
