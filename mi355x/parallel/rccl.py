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
UID_BYTES = 128  # sizeof(ncclUniqueId), asserted against the extension


def _bind_server(addr: str, port: int, world: int) -> socket.socket:
    """Bind the rendezvous listen socket SYNCHRONOUSLY on rank 0 so a port
    conflict raises immediately there (a bind failure inside a background
    thread would leave every other rank timing out with no cause visible)."""
    srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    try:
        srv.bind((addr, port))
    except OSError as e:
        srv.close()
        raise OSError(
            f"RCCL rendezvous: rank 0 could not bind {addr}:{port} "
            f"(MASTER_PORT+{RCCL_PORT_OFFSET}): {e}. Another process is "
            "holding the port — change MASTER_PORT or kill the holder."
        ) from e
    srv.listen(world)
    return srv


def _serve_id(srv: socket.socket, payload: bytes, world: int):
    served = 0
    while served < world - 1:
        conn, _ = srv.accept()
        conn.sendall(payload)
        conn.close()
        served += 1
    srv.close()


def _fetch_id(addr: str, port: int, timeout_s: float | None = None) -> bytes:
    if timeout_s is None:
        timeout_s = float(os.environ.get("MI355X_RDZV_TIMEOUT", "120"))
    deadline = time.time() + timeout_s
    while True:
        try:
            s = socket.create_connection((addr, port), timeout=5)
            buf = b""
            while len(buf) < UID_BYTES:
                chunk = s.recv(UID_BYTES - len(buf))
                if not chunk:
                    break
                buf += chunk
            s.close()
            if len(buf) == UID_BYTES:
                return buf
        except OSError:
            pass
        if time.time() > deadline:
            raise TimeoutError(
                f"RCCL rendezvous: could not fetch ncclUniqueId from "
                f"{addr}:{port} within {timeout_s:.0f}s (override with "
                "MI355X_RDZV_TIMEOUT). Rank 0 is down or unreachable — "
                "check its log for a bind/init error.")
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
                assert len(uid) == UID_BYTES, \
                    f"ncclUniqueId is {len(uid)} bytes, expected {UID_BYTES}"
                srv = _bind_server(master_addr, port, world)
                t = threading.Thread(target=_serve_id,
                                     args=(srv, uid, world), daemon=True)
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
