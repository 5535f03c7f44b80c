"""bench.py's own distributed branch, exercised exactly the way the driver
launches it (python -m torch.distributed.run --nnodes=1 --nproc-per-node N
... bench.py --gpus N), on CPU/gloo at world 2, 4 and 8 — so the round-end
SCALE run's code path (rank/env plumbing, DDP construction, barrier +
max-over-ranks timing, single JSON line from rank 0) is covered before it
ever meets an 8-GPU node."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_bench_dist(world, port, extra=()):
    env = dict(os.environ)
    env.update({"MI355X_BENCH_MODEL": "net"})
    # --standalone: the rendezvous port is OS-assigned at bind time — a
    # fixed --master-port left TIME_WAIT sockets that collided across
    # back-to-back suite runs (the `port` arg is kept but unused)
    cmd = [sys.executable, "-m", "torch.distributed.run", "--standalone",
           "--local-addr", "127.0.0.1",
           "--nproc-per-node", str(world), "bench.py", "--gpus", str(world),
           "--steps", "2", "--warmup", "1", "--batch", "8", "--model", "net",
           *extra]
    r = subprocess.run(cmd, cwd=REPO, env=env, capture_output=True,
                       text=True, timeout=600)
    assert r.returncode == 0, f"stdout:{r.stdout[-2000:]}\nstderr:{r.stderr[-2000:]}"
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected ONE JSON line, got {lines}"
    return json.loads(lines[0])


@pytest.mark.parametrize("world,port", [(2, 29651), (4, 29655), (8, 29663)])
def test_bench_distributed_contract(world, port):
    out = _run_bench_dist(world, port)
    assert out["config"]["parallelism"] == f"dp{world}"
    assert out["config"]["global_batch"] == 8 * world
    assert out["steps"] == 2 and out["warmup"] == 1
    assert out["value"] > 0 and out["ms_per_step"] > 0
    assert out["scaling"] == "weak"
    assert out["data"] == "synthetic"


def test_bench_distributed_fp16_sync_bn():
    """Config-5 flags survive the distributed path on CPU."""
    out = _run_bench_dist(2, 29659, extra=("--fp16", "--sync-bn"))
    assert out["dtype"] == "fp16"
    assert out["config"]["sync_bn"] is True
