"""Multi-process DDP correctness on CPU/gloo, world_size=2:
gradient all-reduce equivalence with single-process large-batch training,
param broadcast at construction, bucket-view gradients."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from ddp_tricks_amd import same_seeds

WORLD = 2


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)


class TinyNet(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = torch.nn.Linear(8, 16)
        self.bn = torch.nn.BatchNorm1d(16)
        self.fc2 = torch.nn.Linear(16, 4)

    def forward(self, x):
        return self.fc2(torch.relu(self.bn(self.fc1(x))))


def _worker_grad_equiv(rank, port, results):
    _init(rank, WORLD, port)
    from ddp_tricks_amd.parallel.ddp import DistributedDataParallel as DDP
    same_seeds(11)
    model = TinyNet()
    ddp = DDP(model, bucket_cap_mb=0.0001)  # force multiple buckets
    torch.manual_seed(123)  # same data on both ranks' generator
    x = torch.randn(WORLD * 4, 8)
    t = torch.randn(WORLD * 4, 4)
    shard_x = x[rank * 4:(rank + 1) * 4]
    shard_t = t[rank * 4:(rank + 1) * 4]
    model.train()
    out = ddp(shard_x)
    loss = ((out - shard_t) ** 2).mean()
    loss.backward()
    ddp.finalize_backward(average=True)
    if rank == 0:
        # serialize by value (numpy) — tensor fd-sharing over the queue races
        # with worker exit and the parent's detach can hit a dead listener
        grads = {k: p.grad.detach().clone().numpy()
                 for k, p in model.named_parameters()}
        results.put(grads)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_ddp_grads_match_fullbatch():
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    port = 29611
    procs = [ctx.Process(target=_worker_grad_equiv, args=(r, port, results))
             for r in range(WORLD)]
    [p.start() for p in procs]
    grads = results.get(timeout=110)
    [p.join(timeout=60) for p in procs]
    assert all(p.exitcode == 0 for p in procs)

    # single-process oracle: mean of shard losses == ... careful: DDP averages
    # gradients of per-shard MEAN losses, which equals the gradient of the
    # mean of the two shard losses.
    same_seeds(11)
    model = TinyNet()
    torch.manual_seed(123)
    x = torch.randn(WORLD * 4, 8)
    t = torch.randn(WORLD * 4, 4)
    model.train()
    loss0 = ((model(x[:4]) - t[:4]) ** 2).mean()
    # fresh BN stats per shard on each rank — replicate rank-local BN by
    # re-running with reset running stats
    same_seeds(11)
    model2 = TinyNet()
    loss1 = ((model2(x[4:]) - t[4:]) ** 2).mean()
    loss = (loss0 + loss1) / 2
    loss.backward()
    # accumulate grads from both replicas
    for (k, p), (k2, p2) in zip(model.named_parameters(), model2.named_parameters()):
        g = (p.grad if p.grad is not None else 0) + (p2.grad if p2.grad is not None else 0)
        assert torch.allclose(torch.from_numpy(grads[k]), g, atol=1e-5), k


def _worker_broadcast(rank, port, results):
    _init(rank, WORLD, port)
    from ddp_tricks_amd.parallel.ddp import DistributedDataParallel as DDP
    torch.manual_seed(100 + rank)  # DIFFERENT init per rank
    model = TinyNet()
    DDP(model)
    if rank == 1:
        results.put({k: v.detach().clone().numpy()
                     for k, v in model.state_dict().items()})
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_param_broadcast_from_rank0():
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    port = 29612
    procs = [ctx.Process(target=_worker_broadcast, args=(r, port, results))
             for r in range(WORLD)]
    [p.start() for p in procs]
    sd1 = results.get(timeout=110)
    [p.join(timeout=60) for p in procs]
    assert all(p.exitcode == 0 for p in procs)
    torch.manual_seed(100)  # rank 0 init
    ref = TinyNet().state_dict()
    for k in ref:
        assert torch.allclose(torch.from_numpy(sd1[k]).float(), ref[k].float()), k


def test_singleproc_ddp_grad_views():
    """world_size=1 path: grads are views into bucket flats."""
    if dist.is_initialized():
        dist.destroy_process_group()
    from ddp_tricks_amd.parallel.ddp import DistributedDataParallel as DDP
    model = TinyNet()
    ddp = DDP(model)
    x = torch.randn(4, 8)
    loss = ddp(x).sum()
    loss.backward()
    ddp.finalize_backward()
    flats = ddp.bucket_flats()
    total = sum(f.numel() for f in flats)
    assert total == sum(p.numel() for p in model.parameters())
    for p in model.parameters():
        assert p.grad is not None
        assert any(p.grad.data_ptr() >= f.data_ptr()
                   and p.grad.data_ptr() < f.data_ptr() + f.numel() * 4
                   for f in flats)


def _worker_eval_buffer_sync(rank, port, results):
    _init(rank, WORLD, port)
    from ddp_tricks_amd.parallel.ddp import DistributedDataParallel as DDP
    torch.manual_seed(11)
    model = TinyNet()
    ddp = DDP(model)
    # diverge the BN running stats rank-locally (post-broadcast updates)
    with torch.no_grad():
        model.bn.running_mean.add_(float(rank + 1))
    ddp.eval()
    with torch.no_grad():
        ddp(torch.zeros(4, 8))   # eval forward must re-sync buffers
    if rank == 1:
        results.put(model.bn.running_mean.numpy().copy())
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_eval_forward_resyncs_buffers():
    """Eval forwards broadcast buffers (rank-identical valid metrics keep
    EarlyStopping in lockstep — SURVEY Appendix A.7)."""
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=_worker_eval_buffer_sync,
                         args=(r, 29613, results)) for r in range(WORLD)]
    [p.start() for p in procs]
    rm1 = results.get(timeout=110)
    [p.join(timeout=60) for p in procs]
    assert all(p.exitcode == 0 for p in procs)
    # rank 1 must hold rank 0's buffers (running_mean += 1.0, not += 2.0)
    assert abs(float(rm1.mean()) - 1.0) < 1e-5, rm1


def _worker_full_stack_sync(rank, port, results):
    """The production wiring (amp O1 + FusedSGD + Lookahead): rank grads and
    params must stay identical after steps — regression for the
    zero_grad(set_to_none=True) bucket-view drop (Lookahead's old default
    detached grads from the bucket flats, so the all-reduce reduced zeros
    and ranks silently desynced; the DDP hook now self-heals re-pointed
    grads and Lookahead defaults to set_to_none=False)."""
    _init(rank, WORLD, port)
    from ddp_tricks_amd import amp
    from ddp_tricks_amd.ops.optim import FusedSGD
    from ddp_tricks_amd.parallel.ddp import DistributedDataParallel as DDP
    from ddp_tricks_amd.utils.lookahead import Lookahead
    amp._state.__init__()
    torch.manual_seed(7)
    model = TinyNet()
    opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9, nesterov=True)
    la = Lookahead(opt, k=2, alpha=0.5)
    model, apex_opt = amp.initialize(model, la, "O1")
    ddp = DDP(model)
    torch.manual_seed(500 + rank)        # DIFFERENT data per rank
    sigs = []
    for step in range(3):
        apex_opt.zero_grad(set_to_none=(step == 1))   # worst case mid-run
        x = torch.randn(4, 8)
        t = torch.randn(4, 4)
        ddp.train()
        loss = ((ddp(x) - t) ** 2).mean()
        with amp.scale_loss(loss, apex_opt) as sl:
            sl.backward()
        gsig = sorted(float(p.grad.double().sum())
                      for p in model.parameters())
        apex_opt.step()
        psig = sorted(float(p.detach().double().sum())
                      for p in model.parameters())
        sigs.append((gsig, psig))
    amp._state.__init__()
    results.put((rank, sigs))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_full_stack_ranks_stay_identical():
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=_worker_full_stack_sync,
                         args=(r, 29614, results)) for r in range(WORLD)]
    [p.start() for p in procs]
    got = dict(results.get(timeout=110) for _ in range(WORLD))
    [p.join(timeout=60) for p in procs]
    assert all(p.exitcode == 0 for p in procs)
    for step in range(3):
        g0, p0 = got[0][step]
        g1, p1 = got[1][step]
        assert g0 == pytest.approx(g1, abs=1e-12), f"grads differ at step {step}"
        assert p0 == pytest.approx(p1, abs=1e-12), f"params differ at step {step}"


def _worker_w4(rank, port, results):
    """World-4 full-stack lockstep (closer to the 8-GPU driver shape):
    bucket count > world size, different data per rank."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=4)
    from ddp_tricks_amd import amp
    from ddp_tricks_amd.ops.optim import FusedSGD
    from ddp_tricks_amd.parallel.ddp import DistributedDataParallel as DDP
    from ddp_tricks_amd.utils.lookahead import Lookahead
    amp._state.__init__()
    torch.manual_seed(7)
    model = TinyNet()
    opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9, nesterov=True)
    la = Lookahead(opt, k=2, alpha=0.5)
    model, apex_opt = amp.initialize(model, la, "O1")
    ddp = DDP(model, bucket_cap_mb=0.0001)
    torch.manual_seed(900 + rank)
    sigs = []
    for _ in range(3):
        apex_opt.zero_grad()
        x = torch.randn(4, 8)
        t = torch.randn(4, 4)
        ddp.train()
        loss = ((ddp(x) - t) ** 2).mean()
        with amp.scale_loss(loss, apex_opt) as sl:
            sl.backward()
        apex_opt.step()
        sigs.append(sorted(float(p.detach().double().sum())
                           for p in model.parameters()))
    amp._state.__init__()
    results.put((rank, sigs))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_full_stack_world4_lockstep():
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=_worker_w4, args=(r, 29619, results))
             for r in range(4)]
    [p.start() for p in procs]
    got = dict(results.get(timeout=170) for _ in range(4))
    [p.join(timeout=60) for p in procs]
    assert all(p.exitcode == 0 for p in procs)
    for step in range(3):
        base = got[0][step]
        for r in range(1, 4):
            assert got[r][step] == pytest.approx(base, abs=1e-12), \
                f"rank {r} diverged at step {step}"
