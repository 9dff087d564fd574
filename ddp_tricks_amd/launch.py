"""Process launcher — torch.distributed.launch-compatible CLI.

The reference is launched as
``python -m torch.distributed.launch --nproc_per_node=2 --master_port=6666
run.py -n=DDP_warmup`` (reference README.md:25).  This module keeps that
contract: it forks one process per GPU, exports
MASTER_ADDR/MASTER_PORT/RANK/LOCAL_RANK/WORLD_SIZE and appends
``--local_rank=<i>`` to each child's argv (omit with ``--use_env``).

Additions over the reference's launcher: a watchdog that
terminates the whole process group when any rank exits non-zero — a crashed
rank kills the job instead of hanging the others in their next collective.

Usage:  python -m ddp_tricks_amd.launch [--nproc_per_node=N]
        [--master_addr=A] [--master_port=P] [--use_env] script.py args...
"""
from __future__ import annotations

import argparse
import os
import signal
import subprocess
import sys
import time


def parse_args(argv=None):
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument("--nproc_per_node", type=int, default=1)
    parser.add_argument("--nnodes", type=int, default=1)
    parser.add_argument("--node_rank", type=int, default=0)
    parser.add_argument("--master_addr", default="127.0.0.1", type=str)
    parser.add_argument("--master_port", default=29500, type=int)
    parser.add_argument("--use_env", default=False, action="store_true")
    parser.add_argument("training_script", type=str)
    parser.add_argument("training_script_args", nargs=argparse.REMAINDER)
    return parser.parse_args(argv)


def main(argv=None) -> int:
    args = parse_args(argv)
    world_size = args.nnodes * args.nproc_per_node
    procs = []
    base_env = dict(os.environ)
    base_env["MASTER_ADDR"] = args.master_addr
    base_env["MASTER_PORT"] = str(args.master_port)
    base_env["WORLD_SIZE"] = str(world_size)

    for local_rank in range(args.nproc_per_node):
        rank = args.node_rank * args.nproc_per_node + local_rank
        env = dict(base_env)
        env["RANK"] = str(rank)
        env["LOCAL_RANK"] = str(local_rank)
        cmd = [sys.executable, "-u", args.training_script]
        if not args.use_env:
            cmd.append(f"--local_rank={local_rank}")
        cmd.extend(args.training_script_args)
        procs.append(subprocess.Popen(cmd, env=env))

    # Watchdog: first non-zero exit tears down the group.
    exit_code = 0
    try:
        while procs:
            alive = []
            for p in procs:
                ret = p.poll()
                if ret is None:
                    alive.append(p)
                elif ret != 0:
                    exit_code = ret
                    print(f"[ddp_tricks_amd.launch] rank process {p.pid} "
                          f"exited with {ret}; terminating group",
                          file=sys.stderr)
                    for q in procs:
                        if q.poll() is None:
                            q.terminate()
                    for q in procs:
                        try:
                            q.wait(timeout=10)
                        except subprocess.TimeoutExpired:
                            q.kill()
                    return exit_code
            procs = alive
            if procs:
                time.sleep(0.2)
    except KeyboardInterrupt:
        for p in procs:
            if p.poll() is None:
                p.send_signal(signal.SIGINT)
        for p in procs:
            p.wait()
        exit_code = 130
    return exit_code


if __name__ == "__main__":
    sys.exit(main())
