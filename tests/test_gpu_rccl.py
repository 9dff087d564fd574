"""Native RCCL communicator coverage on a real GPU (VERDICT r01
next-round #1).

Tier 1: world-1 RCCL comm — exercises ncclCommInitRank, broadcast,
all-reduce (sum/avg), group windows and the current-HIP-stream plumbing of
``ops/hip/comm.cpp`` with real RCCL calls.

Tier 2: world-2 on ONE device — two processes, both on cuda:0, torch PG on
gloo (rendezvous + fallback transport), ``DDPX_NATIVE_RCCL=1``.  If this
RCCL build accepts two ranks per device the full native path runs; if it
refuses, the DDP ctor's defensive fallback must engage and the run must
still be bitwise rank-lockstep.  Either way the test asserts:
  * params bitwise-identical across ranks after every step of the full
    amp + FusedSGD + Lookahead stack (mirrors tests/test_ddp_gloo.py
    full-stack lockstep), and
  * gradients identical to the ``DDPX_NATIVE_RCCL=0`` torch-collective
    branch run at the same seeds (native path changes transport, not math).
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

WORLD = 2


@pytest.mark.timeout(300)
def test_rccl_world1_collectives():
    from ddp_tricks_amd.ops import load_extension
    ext = load_extension()
    assert ext is not None and hasattr(ext, "rccl_unique_id"), \
        "HIP extension must expose the RCCL bindings on a GPU box"
    uid = ext.rccl_unique_id()
    comm = ext.rccl_comm_init(1, 0, uid)
    dev = torch.device("cuda", 0)
    t = torch.arange(8, dtype=torch.float32, device=dev)
    ext.rccl_broadcast(t, 0, comm)          # world-1 broadcast: identity
    torch.cuda.synchronize()
    assert torch.equal(t, torch.arange(8, dtype=torch.float32, device=dev))
    ext.rccl_group_start()
    ext.rccl_all_reduce(t, comm, False)     # world-1 sum: identity
    ext.rccl_group_end()
    torch.cuda.synchronize()
    assert torch.equal(t, torch.arange(8, dtype=torch.float32, device=dev))
    u = torch.full((4,), 3.0, dtype=torch.bfloat16, device=dev)
    ext.rccl_all_reduce(u, comm, True)      # avg over world 1: identity
    torch.cuda.synchronize()
    assert torch.equal(u, torch.full((4,), 3.0, dtype=torch.bfloat16,
                                     device=dev))
    ext.rccl_comm_destroy(comm)


class TinyNet(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = torch.nn.Linear(8, 16)
        self.bn = torch.nn.BatchNorm1d(16)
        self.fc2 = torch.nn.Linear(16, 4)

    def forward(self, x):
        return self.fc2(torch.relu(self.bn(self.fc1(x))))


def _worker(rank, port, native, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["DDPX_NATIVE_RCCL"] = "1" if native else "0"
    torch.cuda.set_device(0)            # BOTH ranks share device 0
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)
    import warnings

    from ddp_tricks_amd import amp
    from ddp_tricks_amd.ops.optim import FusedSGD
    from ddp_tricks_amd.parallel.ddp import DistributedDataParallel as DDP
    from ddp_tricks_amd.utils.lookahead import Lookahead
    amp._state.__init__()
    torch.manual_seed(7)
    model = TinyNet().cuda()
    opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9, nesterov=True)
    la = Lookahead(opt, k=2, alpha=0.5)
    model, apex_opt = amp.initialize(model, la, "O1")
    with warnings.catch_warnings(record=True):
        warnings.simplefilter("always")
        ddp = DDP(model, bucket_cap_mb=0.0001)   # force several buckets
    used_native = ddp._rccl is not None
    torch.manual_seed(500 + rank)        # DIFFERENT data per rank
    sigs = []
    for step in range(3):
        apex_opt.zero_grad()
        x = torch.randn(4, 8).cuda()
        t = torch.randn(4, 4).cuda()
        ddp.train()
        loss = ((ddp(x) - t) ** 2).mean()
        with amp.scale_loss(loss, apex_opt) as sl:
            sl.backward()
        torch.cuda.synchronize()
        gsig = [float(p.grad.double().sum()) for p in model.parameters()]
        apex_opt.step()
        torch.cuda.synchronize()
        psig = [float(p.detach().double().sum()) for p in model.parameters()]
        sigs.append((gsig, psig))
    amp._state.__init__()
    results.put((rank, used_native, sigs))
    dist.barrier()
    dist.destroy_process_group()


def _run_world2(native, port):
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, port, native, results))
             for r in range(WORLD)]
    [p.start() for p in procs]
    got = {}
    for _ in range(WORLD):
        rank, used_native, sigs = results.get(timeout=240)
        got[rank] = (used_native, sigs)
    [p.join(timeout=120) for p in procs]
    assert all(p.exitcode == 0 for p in procs)
    return got


@pytest.mark.timeout(600)
def test_native_rccl_world2_full_stack_lockstep():
    got_native = _run_world2(native=True, port=29721)
    used = [got_native[r][0] for r in range(WORLD)]
    assert used[0] == used[1], "ranks disagreed on native-vs-fallback path"
    print(f"[rccl-test] native RCCL path engaged: {used[0]}")
    # bitwise rank-lockstep through the engaged path
    for step in range(3):
        g0, p0 = got_native[0][1][step]
        g1, p1 = got_native[1][1][step]
        assert g0 == g1, f"grads differ across ranks at step {step}"
        assert p0 == p1, f"params differ across ranks at step {step}"
    # transport equivalence: same numbers through the torch-collective branch
    got_torch = _run_world2(native=False, port=29722)
    for step in range(3):
        gn, pn = got_native[0][1][step]
        gt, pt = got_torch[0][1][step]
        for a, b in zip(gn, gt):
            assert a == pytest.approx(b, rel=1e-6, abs=1e-9), \
                f"native-vs-torch grad mismatch at step {step}"
        for a, b in zip(pn, pt):
            assert a == pytest.approx(b, rel=1e-6, abs=1e-9), \
                f"native-vs-torch param mismatch at step {step}"


@pytest.mark.timeout(300)
def test_rccl_reducer_stream_pattern_world1():
    """Drive the exact enqueue pattern the DDP reducer uses on the native
    path — compute-stream event → comm-stream wait → grouped
    rccl_all_reduce of bucket flats → main-stream wait — through a real
    RCCL communicator (world 1: transport is a no-op, the stream
    semantics and bindings are not)."""
    from ddp_tricks_amd.ops import load_extension
    ext = load_extension()
    uid = ext.rccl_unique_id()
    comm = ext.rccl_comm_init(1, 0, uid)
    dev = torch.device("cuda", 0)
    torch.cuda.set_device(0)
    comm_stream = torch.cuda.Stream(device=dev)
    flats = [torch.randn(1 << k, device=dev) for k in (10, 14, 8)]
    want = [f.clone() for f in flats]
    # produce on the compute stream (in-place mul), then reduce on the
    # comm stream exactly like _on_grad_ready
    for f, w in zip(flats, want):
        f.mul_(3.0)
        w.mul_(3.0)
        ev = torch.cuda.Event()
        ev.record(torch.cuda.current_stream(dev))
        comm_stream.wait_event(ev)
        with torch.cuda.stream(comm_stream):
            ext.rccl_all_reduce(f, comm, False)
    torch.cuda.current_stream(dev).wait_stream(comm_stream)
    # grouped window over all flats (the broadcast path's shape)
    with torch.cuda.stream(comm_stream):
        ext.rccl_group_start()
        for f in flats:
            ext.rccl_broadcast(f, 0, comm)
        ext.rccl_group_end()
    torch.cuda.current_stream(dev).wait_stream(comm_stream)
    torch.cuda.synchronize()
    for f, w in zip(flats, want):
        assert torch.equal(f, w)
    ext.rccl_comm_destroy(comm)
