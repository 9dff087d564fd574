"""Real-MNIST convergence check.

The reference's only quantitative record is a 500-epoch real-MNIST run
(best valid acc 0.99050 — reference experiment_model/logs/DDP_warmup/).
This environment has no network and the reference tree is missing the
60k train-images blob, so exact golden-curve parity is impossible; what
IS possible with the shipped real data (t10k images + labels,
datasets/MNIST/raw/) is a real-distribution convergence check: train the
full reference recipe (Toy_Net, FusedSGD+Lookahead, amp O1, warmup,
plateau, EarlyStopping, double-normalized CE) on 8k REAL MNIST digits and
validate on the held-out 2k.  A bf16 HIP path that converges here runs
the same kernels/optimizer/scaling as the 60k run would.

Usage:  python tools/real_mnist_check.py [--epochs 60] [--batch-size 1024]
Writes per-epoch metrics to stdout (capture into profiles/).
"""
from __future__ import annotations

import argparse
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
os.environ.setdefault("DDPX_NO_TQDM", "1")

import torch  # noqa: E402

from ddp_tricks_amd import amp, same_seeds  # noqa: E402
from ddp_tricks_amd.models.toy_net import Toy_Net  # noqa: E402
from ddp_tricks_amd.ops.functional import cross_entropy_loss  # noqa: E402
from ddp_tricks_amd.ops.optim import FusedSGD  # noqa: E402
from ddp_tricks_amd.utils.callbacks import EarlyStopping  # noqa: E402
from ddp_tricks_amd.utils.data import MNIST, FastBatchLoader  # noqa: E402
from ddp_tricks_amd.utils.engine import iterate_loader  # noqa: E402
from ddp_tricks_amd.utils.lookahead import Lookahead  # noqa: E402
from ddp_tricks_amd.utils.schedulers import (  # noqa: E402
    ReduceLROnPlateau, WarmupLambdaLR,
)


class _Slice:
    """Tensor-backed dataset slice compatible with FastBatchLoader."""

    def __init__(self, images, labels):
        self.images = images
        self.labels = labels

    def __len__(self):
        return self.labels.shape[0]


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--epochs", type=int, default=60)
    p.add_argument("--batch-size", type=int, default=1024)
    p.add_argument("--data-path", default=os.path.join(REPO, "datasets"))
    p.add_argument("--train-n", type=int, default=8000)
    args = p.parse_args()

    ds = MNIST(root=args.data_path, train=False)   # t10k = the REAL data
    if ds.synthetic:
        print("REAL t10k files not found — nothing to check")
        return 1
    same_seeds(42)
    train = _Slice(ds.images[:args.train_n], ds.labels[:args.train_n])
    valid = _Slice(ds.images[args.train_n:], ds.labels[args.train_n:])
    device = torch.device("cuda:0") if torch.cuda.is_available() \
        else torch.device("cpu")
    pin = device.type == "cuda"
    dev_arg = device if pin else None
    train_loader = FastBatchLoader(train, args.batch_size, pin_memory=pin,
                                   device=dev_arg)
    valid_loader = FastBatchLoader(valid, args.batch_size, pin_memory=pin,
                                   device=dev_arg)

    same_seeds(42)
    model = Toy_Net().to(device)
    opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9, nesterov=True)
    la = Lookahead(opt, k=10, alpha=0.5)
    model, apex_opt = amp.initialize(model, la, "O1")
    warmup_epochs = 10
    sched_wu = WarmupLambdaLR(
        opt, lr_lambda=lambda ep: min(ep / warmup_epochs, 1.0))
    sched_re = ReduceLROnPlateau(opt, mode="min", factor=0.1, patience=6)
    early = EarlyStopping(patience=30, verbose=False)

    best = 0.0
    for epoch in range(args.epochs):
        t0 = time.time()
        model.train()
        tr_loss, tr_acc, lr = iterate_loader(train_loader, model,
                                             cross_entropy_loss, 0,
                                             apex_opt, training=True)
        model.eval()
        with torch.no_grad():
            va_loss, va_acc = iterate_loader(valid_loader, model,
                                             cross_entropy_loss, 0, None,
                                             training=False)
        amp.maybe_sync_scaler()
        best = max(best, va_acc)
        print(f"epoch {epoch:3d}  lr {lr:.4f}  train {tr_loss:.3e}/{tr_acc:.4f}"
              f"  valid {va_loss:.3e}/{va_acc:.4f}  best {best:.4f}"
              f"  {time.time() - t0:.2f}s", flush=True)
        if epoch <= 10:
            sched_wu.step()
        sched_re.step(va_loss)
        early(va_loss)
        if early.early_stop:
            print(f"EarlyStopping at epoch {epoch}")
            break
    print(f"FINAL best_valid_acc {best:.4f} on REAL MNIST "
          f"({args.train_n} train / {len(valid)} valid)")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
