"""run.py — CLI + process-group init entry point.

API-compatible with the reference entry point (reference run.py:7-32):
identical 10 flags and defaults, ``--local_rank`` injected by the launcher
(``python -m ddp_tricks_amd.launch`` or torchrun-style LOCAL_RANK env),
then process-group init over RCCL (backend "nccl" IS RCCL on ROCm) and
``train(args)``.  Falls back to gloo on CPU-only machines and fills in
single-process rendezvous defaults so ``python run.py`` works standalone.
"""
import argparse
import os

import torch
import torch.distributed as dist

from ddp_tricks_amd.utils.train import train


def parse_args():
    parser = argparse.ArgumentParser()
    parser.add_argument("-n", "--exp_name", default="DDP_warmup", type=str,
                        help="name of experiment")
    parser.add_argument("-l", "--learning_rate", default=1e-1, type=float,
                        help="learning rate")
    parser.add_argument("-b", "--batch_size", default=1024, type=int,
                        help="batch size")
    parser.add_argument("-e", "--epochs", default=500, type=int,
                        help="epochs")
    parser.add_argument("-w", "--warmup_epochs", default=10, type=int,
                        help="epochs for warmup")
    parser.add_argument("-t", "--warmup_type", default="linear", type=str,
                        help="warmup type")
    parser.add_argument("-s", "--seed_num", default=42, type=int,
                        help="number of random seed")
    parser.add_argument("-d", "--data_path", default="./datasets/", type=str,
                        help="path of dataset")
    parser.add_argument("-p", "--model_path", default="./experiment_model/",
                        type=str, help="path of model")
    parser.add_argument("--local_rank", type=int, default=None,
                        help="local rank for DistributedDataParallel")
    # extensions beyond the reference's 10-flag contract
    parser.add_argument("--model", default="toy_net", type=str,
                        help="model registry name (toy_net, vgg16, resnet18, "
                             "resnet34, resnet50)")
    parser.add_argument("--dataset", default="mnist", type=str,
                        help="dataset registry name (mnist, cifar10, "
                             "imagenet_synthetic)")
    parser.add_argument("--resume", action="store_true",
                        help="save/load the sidecar resume checkpoint")
    args = parser.parse_args()
    if args.local_rank is None:
        args.local_rank = int(os.environ.get("LOCAL_RANK", 0))
    return args


def main():
    args = parse_args()
    print(f"Running DDP on rank: {args.local_rank}")

    # single-process fallback rendezvous (container hostnames may not resolve
    # — always rendezvous on loopback)
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    os.environ.setdefault("RANK", str(args.local_rank))
    os.environ.setdefault("WORLD_SIZE", "1")

    if torch.cuda.is_available():
        torch.cuda.set_device(args.local_rank)
        backend = "nccl"  # RCCL on ROCm
    else:
        backend = "gloo"
    dist.init_process_group(backend=backend, init_method="env://")
    try:
        train(args)
    finally:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
