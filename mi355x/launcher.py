"""torchrun-equivalent launcher: fork one process per GPU, set the env
contract, watch children.

The env contract is the one the reference DDP script reads
(/root/reference/cifar_example_ddp.py:43-45): RANK, LOCAL_RANK,
WORLD_SIZE, plus MASTER_ADDR/MASTER_PORT (single node, 127.0.0.1). Any
child exiting non-zero => SIGTERM the rest and exit non-zero.

Usage:
    python -m mi355x.launcher --nproc-per-node 8 cifar_example_ddp.py [args...]
"""

from __future__ import annotations

import argparse
import os
import signal
import subprocess
import sys
import time


def launch(nproc: int, script: str, script_args: list[str],
           master_addr: str = "127.0.0.1", master_port: int = 29500) -> int:
    procs: list[subprocess.Popen] = []
    base_env = dict(os.environ)
    base_env["MASTER_ADDR"] = master_addr
    base_env["MASTER_PORT"] = str(master_port)
    base_env["WORLD_SIZE"] = str(nproc)
    base_env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    for rank in range(nproc):
        env = dict(base_env)
        env["RANK"] = str(rank)
        env["LOCAL_RANK"] = str(rank)
        procs.append(subprocess.Popen(
            [sys.executable, script, *script_args], env=env))
    rc = 0
    try:
        alive = set(range(nproc))
        while alive:
            for i in sorted(alive):
                r = procs[i].poll()
                if r is None:
                    continue
                alive.discard(i)
                if r != 0:
                    rc = r
                    for j in sorted(alive):
                        procs[j].send_signal(signal.SIGTERM)
                    deadline = time.time() + 10
                    for j in sorted(alive):
                        try:
                            procs[j].wait(timeout=max(0.1, deadline - time.time()))
                        except subprocess.TimeoutExpired:
                            procs[j].kill()
                    return rc
            time.sleep(0.2)
    except KeyboardInterrupt:
        for p in procs:
            if p.poll() is None:
                p.send_signal(signal.SIGTERM)
        rc = 130
    return rc


def main(argv=None):
    ap = argparse.ArgumentParser(description="mi355x fork-per-GPU launcher")
    ap.add_argument("--nproc-per-node", "--nproc_per_node", type=int,
                    dest="nproc", default=None,
                    help="processes to launch (default: visible GPU count)")
    ap.add_argument("--master-addr", default="127.0.0.1")
    ap.add_argument("--master-port", type=int, default=29500)
    ap.add_argument("script")
    ap.add_argument("script_args", nargs=argparse.REMAINDER)
    args = ap.parse_args(argv)
    nproc = args.nproc
    if nproc is None:
        try:
            import torch
            nproc = max(torch.cuda.device_count(), 1)
        except Exception:
            nproc = 1
    sys.exit(launch(nproc, args.script, args.script_args,
                    args.master_addr, args.master_port))


if __name__ == "__main__":
    main()
