"""Scheduler robustness against malformed/hostile clients (no GPU).

The daemon shares a node between tenants; a misbehaving container must
not be able to crash or wedge it.
"""

from __future__ import annotations

import os
import random
import socket
import time

from nvshare_amd import ctl, proto


def raw_conn(sock_dir):
    s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    s.settimeout(5)
    s.connect(proto.scheduler_path(sock_dir))
    return s


def alive(sock_dir):
    st = ctl.status(sock_dir)
    return st is not None


def test_garbage_bytes(sched, sock_dir):
    rng = random.Random(7)
    for _ in range(5):
        s = raw_conn(sock_dir)
        s.sendall(bytes(rng.randrange(256)
                        for _ in range(proto.MSG_SIZE)))
        s.close()
    time.sleep(0.2)
    assert alive(sock_dir)


def test_partial_frame_disconnect(sched, sock_dir):
    s = raw_conn(sock_dir)
    s.sendall(b"\x01" + b"x" * 100)  # 101 of 537 bytes
    s.close()
    time.sleep(0.2)
    assert alive(sock_dir)
    # And a partial frame kept open while others work normally.
    s = raw_conn(sock_dir)
    s.sendall(b"\x04" + b"y" * 10)
    c = proto.Client(sock_dir=sock_dir, pod_name="ok").connect()
    c.register()
    c.send(proto.REQ_LOCK)
    assert c.recv(5).type == proto.LOCK_OK
    c.close()
    s.close()


def test_unknown_message_types(sched, sock_dir):
    for t in (0, 42, 200, 255):
        s = raw_conn(sock_dir)
        m = proto.Message(proto.REGISTER)
        raw = bytearray(m.pack())
        raw[0] = t
        s.sendall(bytes(raw))
        s.close()
    time.sleep(0.2)
    assert alive(sock_dir)


def test_flooding_client(sched, sock_dir):
    """A client spamming REQ_LOCK/RELEASED without reading replies
    cannot wedge the daemon; the daemon may evict it (strict eviction,
    like the reference on partial sends) but must keep serving."""
    c = proto.Client(sock_dir=sock_dir, pod_name="flood").connect()
    c.register()
    try:
        for _ in range(500):
            c.send(proto.REQ_LOCK)
            c.send(proto.LOCK_RELEASED)
    except (BrokenPipeError, ConnectionError):
        pass  # evicted mid-flood: acceptable
    try:
        while True:
            c.recv(0.5)
    except Exception:
        pass
    assert alive(sock_dir)
    c.close()
    # A well-behaved client still gets the full lifecycle afterwards.
    ok = proto.Client(sock_dir=sock_dir, pod_name="after").connect()
    ok.register()
    ok.send(proto.REQ_LOCK)
    assert ok.recv(5).type == proto.LOCK_OK
    ok.close()


def test_many_connections(sched, sock_dir):
    clients = []
    for i in range(60):
        c = proto.Client(sock_dir=sock_dir, pod_name=f"m{i}").connect()
        c.register()
        clients.append(c)
    st = ctl.status(sock_dir)
    assert st.clients == 60
    for c in clients:
        c.close()
    time.sleep(0.5)
    assert ctl.status(sock_dir).clients == 0
