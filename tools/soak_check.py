"""Sustained-run stability check: N bench-style steps, reporting
throughput in windows plus peak device memory (no growth = no leak)."""
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
os.environ.setdefault("DDPX_NO_TQDM", "1")

import torch

from ddp_tricks_amd import amp, same_seeds
from ddp_tricks_amd.models import build_model
from ddp_tricks_amd.ops.functional import cross_entropy_loss
from ddp_tricks_amd.ops.optim import FusedSGD
from ddp_tricks_amd.utils.lookahead import Lookahead

steps = int(sys.argv[1]) if len(sys.argv) > 1 else 10000
device = torch.device("cuda:0")
same_seeds(42)
model = build_model("toy_net").to(device)
opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9, nesterov=True)
la = Lookahead(opt, k=10, alpha=0.5)
model, apex_opt = amp.initialize(model, la, "O1")
x = torch.rand(1024, 1, 28, 28, device=device)
t = torch.randint(0, 10, (1024,), device=device)
model.train()
win = steps // 5
t0 = time.perf_counter()
for i in range(steps):
    apex_opt.zero_grad()
    out = model(x)
    loss = cross_entropy_loss(out, t) / out.shape[0]
    with amp.scale_loss(loss, apex_opt) as sl:
        sl.backward()
    apex_opt.step()
    if (i + 1) % win == 0:
        torch.cuda.synchronize()
        t1 = time.perf_counter()
        print(f"steps {i + 1 - win:5d}-{i + 1:5d}: "
              f"{win * 1024 / (t1 - t0):9.0f} img/s   peak_alloc "
              f"{torch.cuda.max_memory_allocated() / 2**30:.2f} GiB   "
              f"reserved {torch.cuda.memory_reserved() / 2**30:.2f} GiB",
              flush=True)
        t0 = t1
amp.maybe_sync_scaler()
print(f"final loss {float(loss):.3e}  scale {amp._state.scaler.scale}")
