"""train(args) — full training orchestration.

Re-implements the reference trainer's exact wiring (reference
utils/train.py:23-118) on the MI355X-native stack: our MNIST pipeline +
DistributedSampler, our Toy_Net (HIP kernels on GPU), FusedSGD + Lookahead,
our amp runtime, our RCCL-bucketed DDP, our tfevents writer, identical
epoch loop / scheduler gating / early stopping / best-checkpoint rule.

Behavioral quirks preserved: epoch 0 trains at LR 0;
warmup steps while ``epoch <= warmup_epochs``; cosine branch freezes after
warmup; plateau patience 6 / factor 0.1 on valid loss; EarlyStopping
patience 30; unsharded validation on every rank; train metrics rank-local;
rank-0 best-valid-acc gated ``torch.save(module.state_dict())`` to
``{model_path}/{exp_name}.pt`` (37-key fp32 layout).
"""
from __future__ import annotations

import math
import os
import time

import torch
from torch.utils.data import DataLoader

from .. import amp
from ..models import build_model
from ..ops.optim import FusedSGD
from ..parallel.ddp import DistributedDataParallel as DDP
from .callbacks import EarlyStopping, same_seeds
from .data import DATASETS, DistributedSampler, CudaPrefetcher, FastBatchLoader
from .engine import iterate_loader
from .lookahead import Lookahead
from .schedulers import ReduceLROnPlateau, WarmupLambdaLR
from .tboard import SummaryWriter


def _resolve_device(local_rank):
    if torch.cuda.is_available():
        rank = local_rank if local_rank is not None and local_rank >= 0 else 0
        torch.cuda.set_device(rank)
        return torch.device("cuda", rank)
    return torch.device("cpu")


def train(args):
    local_rank = getattr(args, "local_rank", 0) or 0
    device = _resolve_device(local_rank)

    # Extensions over the reference's 10-flag contract:
    # --model / --dataset select from the registries; --resume restarts
    # from the sidecar checkpoint.  Defaults reproduce the reference.
    model_name = getattr(args, "model", None) or "toy_net"
    dataset_name = getattr(args, "dataset", None) or "mnist"
    model_kwargs = dict(getattr(args, "model_kwargs", None) or {})
    if model_name.startswith("resnet") and dataset_name == "cifar10":
        model_kwargs.setdefault("num_classes", 10)
        model_kwargs.setdefault("cifar_stem", True)
    if model_name.startswith("vgg") and dataset_name == "cifar10":
        model_kwargs.setdefault("num_classes", 10)
        model_kwargs.setdefault("cifar_head", True)
    Dataset = DATASETS[dataset_name]

    # Data pipeline (reference utils/train.py:24-30: sharded train loader,
    # UNSHARDED valid loader evaluated in full on every rank)
    train_set = Dataset(root=args.data_path, train=True, download=True)
    train_sampler = DistributedSampler(train_set)
    same_seeds(args.seed_num)
    pin = device.type == "cuda"
    dev_arg = device if pin else None
    if hasattr(train_set, "images"):   # in-memory tensors: vectorized path
        train_loader = FastBatchLoader(train_set, args.batch_size,
                                       sampler=train_sampler, pin_memory=pin,
                                       device=dev_arg)
    else:
        train_loader = DataLoader(train_set, batch_size=args.batch_size,
                                  shuffle=False, pin_memory=pin,
                                  sampler=train_sampler)
    valid_set = Dataset(root=args.data_path, train=False, download=True)
    if getattr(train_set, "synthetic", False) \
            and not getattr(valid_set, "synthetic", True):
        # Coherence guard: the train images are absent (synthetic stand-in)
        # but real valid files exist — validating a synthetic-trained model
        # on real digits reports chance accuracy and would silently distort
        # EarlyStopping/plateau/best-checkpoint decisions.  Keep the pair
        # learnable: synthetic valid drawn from the same class templates.
        print("[data] train split is synthetic but real valid files exist; "
              "using the synthetic valid split for a coherent train/valid "
              "pair")
        valid_set = Dataset(root=args.data_path, train=False, download=False,
                            synthetic=True)
    if hasattr(valid_set, "images"):
        valid_loader = FastBatchLoader(valid_set, args.batch_size,
                                       pin_memory=pin, device=dev_arg)
    else:
        valid_loader = DataLoader(valid_set, batch_size=args.batch_size,
                                  shuffle=False, pin_memory=pin)
    if device.type == "cuda":
        # FastBatchLoader already yields device tensors with its own
        # one-ahead copy stream; only wrap generic DataLoaders
        if not isinstance(train_loader, FastBatchLoader):
            train_loader = CudaPrefetcher(train_loader, device)
        if not isinstance(valid_loader, FastBatchLoader):
            valid_loader = CudaPrefetcher(valid_loader, device)

    print(f"Now Training: {args.exp_name}")

    # Model (seeded identically on all ranks before the DDP broadcast —
    # reference utils/train.py:34-36, README rationale)
    same_seeds(args.seed_num)
    model = build_model(model_name, **model_kwargs)
    model = model.to(device)

    os.makedirs(os.path.join(args.model_path, "logs"), exist_ok=True)
    latest_model_path = os.path.join(args.model_path, args.exp_name)
    optimizer = FusedSGD(model.parameters(), lr=args.learning_rate,
                         momentum=0.9, nesterov=True)
    lookahead = Lookahead(optimizer=optimizer, k=10, alpha=0.5)
    from ..ops.functional import cross_entropy_loss as loss_function
    best_valid_acc = 0

    # LR warmup lambda (reference utils/train.py:48-51, incl. the dead
    # cosine decay quirk — the caller only steps warmup while
    # epoch <= warmup_epochs)
    if args.warmup_type == "linear":
        def warm_up(epoch):
            return epoch / args.warmup_epochs if epoch <= args.warmup_epochs else 1
    elif args.warmup_type == "cosine":
        def warm_up(epoch):
            if epoch <= args.warmup_epochs:
                return epoch / args.warmup_epochs
            return 0.5 * (math.cos((epoch - args.warmup_epochs)
                                   / (args.epochs - args.warmup_epochs) * math.pi) + 1)
    else:
        raise ValueError(f"unknown warmup_type {args.warmup_type!r}")
    scheduler_wu = WarmupLambdaLR(optimizer=optimizer, lr_lambda=warm_up)
    scheduler_re = ReduceLROnPlateau(optimizer=optimizer, mode="min",
                                     factor=0.1, patience=6, verbose=True)
    early_stopping = EarlyStopping(patience=30, verbose=True)

    # amp before DDP wrap (required order — SURVEY Appendix A.12)
    model, apex_optimizer = amp.initialize(model, optimizers=lookahead,
                                           opt_level="O1")
    parallel_model = DDP(model)

    # Resume from the sidecar checkpoint (extension; the reference saves
    # best weights only and has no resume path — SURVEY §5.4).  Every rank
    # loads the same file, so state stays rank-consistent.
    start_epoch = 0
    resume_path = f"{latest_model_path}.resume.pt"
    resume_on = bool(getattr(args, "resume", False))
    if resume_on and os.path.exists(resume_path):
        ck = torch.load(resume_path, map_location=device, weights_only=False)
        parallel_model.module.load_state_dict(ck["model"])
        apex_optimizer.load_state_dict(ck["optimizer"])
        scheduler_wu.load_state_dict(ck["scheduler_wu"])
        scheduler_re.load_state_dict(ck["scheduler_re"])
        early_stopping.load_state_dict(ck["early_stopping"])
        amp.load_state_dict(ck.get("amp", {}))
        best_valid_acc = ck["best_valid_acc"]
        start_epoch = ck["epoch"] + 1
        from ..ops.functional import clear_weight_cache
        clear_weight_cache()
        print(f"resumed {args.exp_name} at epoch {start_epoch}")

    if local_rank == 0:
        tb = SummaryWriter(os.path.join(args.model_path, "logs", args.exp_name))

    for epoch in range(start_epoch, args.epochs):
        epoch_start_time = time.time()
        train_sampler.set_epoch(epoch)

        parallel_model.train()
        train_loss, train_acc, curr_lr = iterate_loader(
            loader=train_loader, model=parallel_model,
            loss_function=loss_function, local_rank=local_rank,
            apex_optimizer=apex_optimizer, training=True)

        parallel_model.eval()
        with torch.no_grad():
            valid_loss, valid_acc = iterate_loader(
                loader=valid_loader, model=parallel_model,
                loss_function=loss_function, local_rank=local_rank,
                apex_optimizer=None, training=False)

        if local_rank == 0:
            tb.add_scalar("LR", curr_lr, epoch)
            tb.add_scalars("Loss", {"train": train_loss, "valid": valid_loss}, epoch)
            tb.add_scalars("Acc", {"train": train_acc, "valid": valid_acc}, epoch)

        phase_info = ""
        # DDPX_PHASE_TIMERS=1; the engine caches PhaseTimers
        # per device (events/streams are device-bound)
        tcache = getattr(iterate_loader, "_timers", None) or {}
        tm = tcache.get(device)
        if tm is not None:
            phase_info = f", phases[{tm.format()}]"
            tm.reset()
        print(f"epoch: {epoch:03d}/{args.epochs}, "
              f"time: {time.time() - epoch_start_time:.2f}s, "
              f"learning_rate: {curr_lr}, "
              f"train_loss: {train_loss:.4f}, train_acc: {train_acc:.4f}, "
              f"valid_loss: {valid_loss:.4f}, valid_acc: {valid_acc:.4f}"
              f"{phase_info}")

        # scheduler gating exactly as the reference (utils/train.py:104-106)
        if epoch <= args.warmup_epochs:
            scheduler_wu.step()
        scheduler_re.step(valid_loss)
        early_stopping(valid_loss)
        if early_stopping.early_stop:
            break

        if local_rank == 0 and valid_acc > best_valid_acc:
            best_valid_acc = valid_acc
            torch.save(parallel_model.module.state_dict(),
                       f"{latest_model_path}.pt")

        if local_rank == 0 and resume_on:
            torch.save({
                "epoch": epoch,
                "model": parallel_model.module.state_dict(),
                "optimizer": apex_optimizer.state_dict(),
                "scheduler_wu": scheduler_wu.state_dict(),
                "scheduler_re": scheduler_re.state_dict(),
                "early_stopping": early_stopping.state_dict(),
                "amp": amp.state_dict(),
                "best_valid_acc": best_valid_acc,
            }, resume_path)

    if local_rank == 0:
        tb.close()
