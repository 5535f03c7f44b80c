"""Python side of the native RCCL communicator.

TCP rendezvous: rank 0 ncclGetUniqueId()s and serves the 128-byte id on
MASTER_ADDR:(MASTER_PORT+RCCL_PORT_OFFSET); other ranks connect (with
retry) and read it; then every rank calls ncclCommInitRank — the same
bootstrap shape as ProcessGroupNCCL's TCPStore exchange, ~60 lines
instead of a store abstraction. Collectives run on a dedicated
high-priority HIP stream (mi355x/csrc/rccl_comm.cpp) so DDP bucket
all-reduces overlap backward compute.
"""

from __future__ import annotations

import os
import socket
import time

RCCL_PORT_OFFSET = 17


def _serve_id(addr: str, port: int, payload: bytes, world: int):
    srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    srv.bind((addr, port))
    srv.listen(world)
    served = 0
    while served < world - 1:
        conn, _ = srv.accept()
        conn.sendall(payload)
        conn.close()
        served += 1
    srv.close()


def _fetch_id(addr: str, port: int, timeout_s: float = 120.0) -> bytes:
    deadline = time.time() + timeout_s
    while True:
        try:
            s = socket.create_connection((addr, port), timeout=5)
            buf = b""
            while len(buf) < 128:
                chunk = s.recv(128 - len(buf))
                if not chunk:
                    break
                buf += chunk
            s.close()
            if len(buf) == 128:
                return buf
        except OSError:
            pass
        if time.time() > deadline:
            raise TimeoutError("RCCL rendezvous: could not fetch ncclUniqueId")
        time.sleep(0.2)


class NativeComm:
    """Our RCCL communicator over xGMI. One per process (per GPU)."""

    def __init__(self, rank: int | None = None, world: int | None = None,
                 master_addr: str | None = None, port: int | None = None):
        from mi355x.ops import ext
        import threading

        rank = int(os.environ.get("RANK", "0")) if rank is None else rank
        world = int(os.environ.get("WORLD_SIZE", "1")) if world is None else world
        master_addr = master_addr or os.environ.get("MASTER_ADDR", "127.0.0.1")
        port = port or int(os.environ.get("MASTER_PORT", "29500")) + RCCL_PORT_OFFSET
        self.rank, self.world = rank, world
        if world > 1:
            if rank == 0:
                uid = ext().rccl_get_unique_id()
                t = threading.Thread(target=_serve_id,
                                     args=(master_addr, port, uid, world),
                                     daemon=True)
                t.start()
            else:
                uid = _fetch_id(master_addr, port)
        else:
            uid = ext().rccl_get_unique_id()
        self._c = ext().RcclComm(rank, world, uid)

    def all_reduce(self, t):
        self._c.all_reduce(t)

    def broadcast(self, t, root=0):
        self._c.broadcast(t, root)

    def all_gather(self, out, inp):
        self._c.all_gather(out, inp)

    def reduce_scatter(self, out, inp):
        self._c.reduce_scatter(out, inp)

    def wait(self):
        self._c.wait()

    def barrier(self):
        self._c.barrier()


_native: NativeComm | None = None


def native_comm() -> NativeComm:
    global _native
    if _native is None:
        _native = NativeComm()
    return _native
