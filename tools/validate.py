"""Evaluate a saved best-weights checkpoint on the validation set.

Usage:  python tools/validate.py -p ./experiment_model -n DDP_warmup \
            [--model toy_net] [--dataset mnist] [-b 1024] [-d ./datasets/]

Loads ``{model_path}/{exp_name}.pt`` (the reference's rank-0 best-checkpoint
layout, reference utils/train.py:115), runs the full validation set through
the HIP path (or CPU fallback), and prints loss/accuracy — the counterpart
to the trainer's per-epoch valid numbers.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
from torch.utils.data import DataLoader  # noqa: E402

from ddp_tricks_amd.models import build_model  # noqa: E402
from ddp_tricks_amd.ops.functional import (argmax_correct,  # noqa: E402
                                           cross_entropy_loss)
from ddp_tricks_amd.utils.data import DATASETS  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("-n", "--exp_name", default="DDP_warmup")
    ap.add_argument("-p", "--model_path", default="./experiment_model/")
    ap.add_argument("-d", "--data_path", default="./datasets/")
    ap.add_argument("-b", "--batch_size", type=int, default=1024)
    ap.add_argument("--model", default="toy_net")
    ap.add_argument("--dataset", default="mnist")
    args = ap.parse_args()

    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    kwargs = {}
    if args.model.startswith("resnet") and args.dataset == "cifar10":
        kwargs = {"num_classes": 10, "cifar_stem": True}
    model = build_model(args.model, **kwargs).to(device)
    ckpt = os.path.join(args.model_path, f"{args.exp_name}.pt")
    if not os.path.exists(ckpt):
        raise SystemExit(
            f"validate.py: no checkpoint at {ckpt} — train first "
            f"(run.py -n={args.exp_name} -p={args.model_path}) or pass "
            f"-n/-p for an existing experiment")
    sd = torch.load(ckpt, map_location=device, weights_only=True)
    model.load_state_dict(sd)
    model.eval()

    ds = DATASETS[args.dataset](root=args.data_path, train=False)
    loader = DataLoader(ds, batch_size=args.batch_size, shuffle=False,
                        pin_memory=device.type == "cuda")
    loss_sum = torch.zeros((), dtype=torch.float64, device=device)
    correct = torch.zeros((), dtype=torch.long, device=device)
    n = 0
    with torch.no_grad():
        for image, target in loader:
            image = image.to(device, non_blocking=True)
            target = target.to(device, dtype=torch.long, non_blocking=True)
            out = model(image)
            batch_loss = cross_entropy_loss(out, target) / out.shape[0]
            loss_sum += batch_loss.detach().double() * image.shape[0]
            correct += argmax_correct(out.detach(), target)
            n += image.shape[0]
    print(f"checkpoint: {ckpt}")
    print(f"valid_loss: {float(loss_sum / n):.6f}, "
          f"valid_acc: {float(correct.double() / n):.4f}, n={n}")


if __name__ == "__main__":
    main()
