"""DistributedDataParallel — bucketed gradient all-reduce over RCCL/xGMI,
overlapped with backward on a side HIP stream.

A from-scratch replacement for the torch C++ reducer the reference leans on
(DDP wrap at reference utils/train.py:61; reducer semantics SURVEY N2/N3):

  * param broadcast from rank 0 at construction (K2) and per-forward buffer
    broadcast (K3, ``broadcast_buffers=True`` default) — buffers are
    coalesced once into two flat tensors (fp32 / int64) and the module's
    buffer entries re-pointed at views, so the per-forward broadcast is two
    small collectives instead of thirteen;
  * gradients live as VIEWS into per-bucket flat fp32 buffers — autograd
    accumulates straight into the bucket, no pack/copy pass;
  * buckets are assigned in reverse registration order (≈ backward
    completion order) and each bucket's SUM all-reduce launches on a
    dedicated comm stream the moment its last gradient lands
    (post-accumulate hooks), overlapping communication with the rest of
    backward;
  * bucket size defaults to 4 MiB: xGMI is 7 point-to-point links per GPU
    (≈153 GB/s each), so many medium shards keep all links busy — not the
    CUDA-era 1/25 MiB split; override with
    ``bucket_cap_mb`` or env DDPX_BUCKET_MB;
  * the 1/world_size average is NOT applied here — it is folded into the
    amp unscale pass (one multiplier, one kernel; see amp.scale_loss), or
    applied by ``finalize_backward(average=True)`` when amp is off.

Works with backend "nccl" (= RCCL on ROCm) on GPU and "gloo" on CPU
(the multi-process CPU test tier).
"""
from __future__ import annotations

import os
from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from .. import amp as amp_mod


class _Bucket:
    __slots__ = ("params", "flat", "views", "ready", "expected", "work", "event")

    def __init__(self):
        self.params: List[torch.nn.Parameter] = []
        self.flat: Optional[torch.Tensor] = None
        self.views = []
        self.ready = 0
        self.expected = 0
        self.work = None
        self.event = None


class DistributedDataParallel(nn.Module):
    def __init__(self, module: nn.Module, device_ids=None, output_device=None,
                 broadcast_buffers: bool = True, bucket_cap_mb: float = None,
                 process_group=None):
        super().__init__()
        self.module = module
        self.process_group = process_group
        self.broadcast_buffers = broadcast_buffers
        self.world_size = dist.get_world_size(process_group) if dist.is_initialized() else 1
        self.rank = dist.get_rank(process_group) if dist.is_initialized() else 0
        if bucket_cap_mb is None:
            bucket_cap_mb = float(os.environ.get("DDPX_BUCKET_MB", "4"))
        self._bucket_cap = int(bucket_cap_mb * 1024 * 1024)
        self._device = next(module.parameters()).device
        self._is_cuda = self._device.type == "cuda"
        self._comm_stream = torch.cuda.Stream(device=self._device) if self._is_cuda else None

        self._params = [p for p in module.parameters() if p.requires_grad]
        self._buckets: List[_Bucket] = []
        self._param_bucket = {}
        self._launched: List[_Bucket] = []

        # Native RCCL communicator: rendezvous the ncclUniqueId
        # over the torch TCP store, then talk to RCCL directly — bucket
        # all-reduces and buffer broadcasts bypass ProcessGroupNCCL.
        self._rccl = None
        self._ext = None
        if self._is_cuda and self.world_size > 1 \
                and os.environ.get("DDPX_NATIVE_RCCL", "1") == "1":
            from ..ops import load_extension
            ext = load_extension()
            if ext is not None and hasattr(ext, "rccl_unique_id"):
                # Defensive: if RCCL refuses the topology (e.g. two ranks
                # sharing one device in a test rig), fall back to the torch
                # process-group path rather than dying — every rank must
                # take the SAME branch, so the failure flag is all-reduced
                # before committing to the native path.
                err = None
                try:
                    obj = [ext.rccl_unique_id() if self.rank == 0 else None]
                    dist.broadcast_object_list(obj, src=0,
                                               group=self.process_group)
                    rccl = ext.rccl_comm_init(self.world_size, self.rank,
                                              obj[0])
                except Exception as e:  # noqa: BLE001 — any init failure
                    err = e
                    rccl = None
                flags = [None] * self.world_size
                dist.all_gather_object(flags, 0 if err is None else 1,
                                       group=self.process_group)
                if sum(flags) == 0:
                    self._rccl = rccl
                    self._ext = ext
                else:
                    if rccl is not None:
                        ext.rccl_comm_destroy(rccl)
                    if self.rank == 0:
                        import warnings
                        warnings.warn(
                            f"native RCCL init failed ({err}); falling back "
                            "to the torch process-group collectives")

        if self.world_size > 1:
            self._broadcast_params()
        self._coalesce_buffers()
        self._build_buckets()
        self._register_hooks()
        if amp_mod.is_enabled():
            amp_mod.register_ddp(self)

    # ------------------------------------------------------------- setup ---

    def _broadcast_params(self) -> None:
        with torch.no_grad():
            if self._rccl is not None:
                self._ext.rccl_group_start()
                for p in self.module.parameters():
                    self._ext.rccl_broadcast(p.data, 0, self._rccl)
                for b in self.module.buffers():
                    self._ext.rccl_broadcast(b.data, 0, self._rccl)
                self._ext.rccl_group_end()
                torch.cuda.synchronize(self._device)
                return
            for p in self.module.parameters():
                dist.broadcast(p.data, src=0, group=self.process_group)
            for b in self.module.buffers():
                dist.broadcast(b.data, src=0, group=self.process_group)

    def _coalesce_buffers(self) -> None:
        """Pack module buffers into flat fp32/int64 tensors; re-point the
        module's buffer registry at views so in-place running-stat updates
        land in the flats and the per-forward broadcast is 2 collectives."""
        self._buf_flats = []
        by_dtype = {}
        entries = []  # (submodule, name, tensor)
        for mod in self.module.modules():
            for name, buf in list(mod._buffers.items()):
                if buf is None:
                    continue
                entries.append((mod, name, buf))
                by_dtype.setdefault(buf.dtype, []).append(buf.numel())
        offsets = {}
        for dtype, sizes in by_dtype.items():
            flat = torch.empty(sum(sizes), dtype=dtype, device=self._device)
            self._buf_flats.append(flat)
            offsets[dtype] = [flat, 0]
        for mod, name, buf in entries:
            flat, off = offsets[buf.dtype]
            view = flat.narrow(0, off, buf.numel()).view_as(buf)
            view.copy_(buf)
            mod._buffers[name] = view
            offsets[buf.dtype][1] = off + buf.numel()

    def _build_buckets(self) -> None:
        bucket = _Bucket()
        size = 0
        # reverse registration order ≈ backward completion order
        for p in reversed(self._params):
            nbytes = p.numel() * p.element_size()
            if bucket.params and size + nbytes > self._bucket_cap:
                self._buckets.append(bucket)
                bucket, size = _Bucket(), 0
            bucket.params.append(p)
            size += nbytes
        if bucket.params:
            self._buckets.append(bucket)
        self._param_view = {}
        for b in self._buckets:
            total = sum(p.numel() for p in b.params)
            b.flat = torch.zeros(total, dtype=torch.float32, device=self._device)
            off = 0
            for p in b.params:
                view = b.flat.narrow(0, off, p.numel()).view_as(p)
                p.grad = view
                b.views.append(view)
                self._param_view[p] = view
                off += p.numel()
            b.expected = len(b.params)
            self._param_bucket.update({p: b for p in b.params})

    def _register_hooks(self) -> None:
        for p in self._params:
            p.register_post_accumulate_grad_hook(self._on_grad_ready)

    # ----------------------------------------------------------- runtime ---

    def _on_grad_ready(self, p: torch.nn.Parameter) -> None:
        # Self-heal: if something dropped the bucket view (a
        # zero_grad(set_to_none=True) from user code), autograd accumulated
        # this step's gradient into a FRESH tensor — fold it back into the
        # bucket and re-point, or the all-reduce would reduce zeros.
        view = self._param_view[p]
        if p.grad is not view:
            with torch.no_grad():
                view.copy_(p.grad)
            p.grad = view
        b = self._param_bucket[p]
        b.ready += 1
        if b.ready < b.expected:
            return
        b.ready = 0
        if self.world_size <= 1:
            return
        if self._is_cuda:
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream(self._device))
            self._comm_stream.wait_event(ev)
            with torch.cuda.stream(self._comm_stream):
                if self._rccl is not None:
                    self._ext.rccl_all_reduce(b.flat, self._rccl, False)
                    b.work = None
                else:
                    b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                             group=self.process_group,
                                             async_op=True)
        else:
            b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                     group=self.process_group, async_op=True)
        self._launched.append(b)

    def finalize_backward(self, average: bool = True) -> None:
        """Wait for in-flight bucket reductions; optionally apply the
        1/world_size average (skipped when amp folds it into unscale)."""
        for b in self._launched:
            if b.work is not None:
                b.work.wait()  # for nccl: enqueues stream dependency only
                b.work = None
        if self._is_cuda and self.world_size > 1:
            torch.cuda.current_stream(self._device).wait_stream(self._comm_stream)
        self._launched.clear()
        if average and self.world_size > 1:
            torch._foreach_mul_([b.flat for b in self._buckets],
                                1.0 / self.world_size)

    def bucket_flats(self) -> List[torch.Tensor]:
        return [b.flat for b in self._buckets]

    def zero_grad_buckets(self) -> None:
        torch._foreach_zero_([b.flat for b in self._buckets])

    def forward(self, *args, **kwargs):
        # NOTE: broadcast in EVAL forwards too (torch-DDP semantics): ranks'
        # BN running stats drift apart during training (rank-local updates
        # land after each pre-forward sync), and rank-identical validation
        # metrics are what keep the un-collectivized EarlyStopping/plateau
        # decisions in lockstep.
        if (self.broadcast_buffers and self.world_size > 1
                and self._buf_flats):
            if self._rccl is not None:
                self._ext.rccl_group_start()
                for flat in self._buf_flats:
                    self._ext.rccl_broadcast(flat, 0, self._rccl)
                self._ext.rccl_group_end()
            else:
                for flat in self._buf_flats:
                    dist.broadcast(flat, src=0, group=self.process_group)
        return self.module(*args, **kwargs)

    # ------------------------------------------------------- passthrough ---

    def state_dict(self, *args, **kwargs):
        return super().state_dict(*args, **kwargs)
