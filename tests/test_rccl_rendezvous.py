"""Socket-level unit tests of the ncclUniqueId TCP rendezvous
(mi355x/parallel/rccl.py) — no GPU needed. The rendezvous is the piece of
the native comm path that runs before any RCCL call, so its failure modes
(port conflict on rank 0, absent rank 0) must be loud and attributable
(make the multi-GPU path bulletproof)."""

import os
import socket
import threading
import time

import pytest

from mi355x.parallel.rccl import (UID_BYTES, _bind_server, _fetch_id,
                                  _serve_id)

ADDR = "127.0.0.1"


def _free_port():
    s = socket.socket()
    s.bind((ADDR, 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_roundtrip_world4():
    """Rank 0 serves one 128-byte uid; three peers each fetch it intact."""
    port = _free_port()
    payload = os.urandom(UID_BYTES)
    srv = _bind_server(ADDR, port, world=4)
    t = threading.Thread(target=_serve_id, args=(srv, payload, 4), daemon=True)
    t.start()
    got = []
    errs = []

    def client():
        try:
            got.append(_fetch_id(ADDR, port, timeout_s=20))
        except Exception as e:
            errs.append(e)

    clients = [threading.Thread(target=client) for _ in range(3)]
    for c in clients:
        c.start()
    for c in clients:
        c.join(timeout=30)
    t.join(timeout=30)
    assert not errs, errs
    assert got == [payload] * 3


def test_fetch_retries_until_server_appears():
    """Peers may connect before rank 0 binds — fetch retries, not fails."""
    port = _free_port()
    payload = os.urandom(UID_BYTES)
    result = {}

    def late_server():
        time.sleep(1.0)
        srv = _bind_server(ADDR, port, world=2)
        _serve_id(srv, payload, 2)

    t = threading.Thread(target=late_server, daemon=True)
    t.start()
    result["uid"] = _fetch_id(ADDR, port, timeout_s=20)
    t.join(timeout=10)
    assert result["uid"] == payload


def test_fetch_timeout_is_attributable():
    port = _free_port()  # nothing listening
    with pytest.raises(TimeoutError) as ei:
        _fetch_id(ADDR, port, timeout_s=1.0)
    msg = str(ei.value)
    assert f"{ADDR}:{port}" in msg          # where it tried
    assert "MI355X_RDZV_TIMEOUT" in msg     # how to extend
    assert "rank 0" in msg.lower()          # whom to suspect


def test_bind_conflict_raises_on_rank0():
    holder = socket.socket()
    holder.bind((ADDR, 0))
    holder.listen(1)  # actively listening: SO_REUSEADDR won't mask this
    port = holder.getsockname()[1]
    try:
        with pytest.raises(OSError) as ei:
            _bind_server(ADDR, port, world=2)
        assert "MASTER_PORT" in str(ei.value)
        assert str(port) in str(ei.value)
    finally:
        holder.close()


def test_serve_exactly_world_minus_one_then_closes():
    """The listen socket closes after world-1 serves — a later run on the
    same port must not read a stale uid from a leftover server."""
    port = _free_port()
    payload = os.urandom(UID_BYTES)
    srv = _bind_server(ADDR, port, world=2)
    t = threading.Thread(target=_serve_id, args=(srv, payload, 2), daemon=True)
    t.start()
    assert _fetch_id(ADDR, port, timeout_s=10) == payload
    t.join(timeout=10)
    assert not t.is_alive()
    with pytest.raises(TimeoutError):
        _fetch_id(ADDR, port, timeout_s=1.0)


def test_roundtrip_world8_burst():
    """8-rank fan-in, the driver's SCALE-run shape: seven peers connect
    SIMULTANEOUSLY (barrier-released) against one rank-0 listen socket
    whose backlog must absorb the burst; every uid arrives intact."""
    port = _free_port()
    payload = os.urandom(UID_BYTES)
    srv = _bind_server(ADDR, port, world=8)
    t = threading.Thread(target=_serve_id, args=(srv, payload, 8), daemon=True)
    t.start()
    got = []
    errs = []
    release = threading.Barrier(7)

    def client():
        try:
            release.wait(timeout=20)
            got.append(_fetch_id(ADDR, port, timeout_s=20))
        except Exception as e:
            errs.append(e)

    clients = [threading.Thread(target=client) for _ in range(7)]
    for c in clients:
        c.start()
    for c in clients:
        c.join(timeout=30)
    t.join(timeout=30)
    assert not errs, errs
    assert got == [payload] * 7
