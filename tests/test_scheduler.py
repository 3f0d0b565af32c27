"""Scheduler daemon behavior tests over the wire protocol (no GPU).

Mock clients drive the full lock lifecycle: FCFS grants, TQ-timer
preemption, early release, SCHED on/off broadcast, SET_TQ, eviction.
This is the protocol/unit harness the reference lacked.
"""

from __future__ import annotations

import time

import pytest

from nvshare_amd import ctl, proto


def make_client(sock_dir, name="c", gpu=0):
    c = proto.Client(sock_dir=sock_dir, pod_name=name, gpu=gpu)
    c.connect()
    c.register()
    return c


def test_register_assigns_ids(sched, sock_dir):
    a = make_client(sock_dir, "a")
    b = make_client(sock_dir, "b")
    assert a.client_id != 0 and b.client_id != 0
    assert a.client_id != b.client_id
    assert a.scheduling_on and b.scheduling_on
    a.close()
    b.close()


def test_lock_grant_and_release(sched, sock_dir):
    a = make_client(sock_dir, "a")
    a.send(proto.REQ_LOCK)
    assert a.recv(5).type == proto.LOCK_OK
    a.send(proto.LOCK_RELEASED)
    # Re-request immediately: must be granted again.
    a.send(proto.REQ_LOCK)
    assert a.recv(5).type == proto.LOCK_OK
    a.close()


def test_fcfs_order(sched, sock_dir):
    clients = [make_client(sock_dir, f"c{i}") for i in range(4)]
    # c0 takes the lock; c1..c3 queue in order.
    clients[0].send(proto.REQ_LOCK)
    assert clients[0].recv(5).type == proto.LOCK_OK
    for c in clients[1:]:
        c.send(proto.REQ_LOCK)
        time.sleep(0.05)  # ensure queue order matches send order
    clients[0].send(proto.LOCK_RELEASED)
    assert clients[1].recv(5).type == proto.LOCK_OK
    clients[1].send(proto.LOCK_RELEASED)
    assert clients[2].recv(5).type == proto.LOCK_OK
    clients[2].send(proto.LOCK_RELEASED)
    assert clients[3].recv(5).type == proto.LOCK_OK
    for c in clients:
        c.close()


def test_tq_preemption(sched, sock_dir):
    """Holder gets DROP_LOCK ~TQ after grant when someone waits."""
    a = make_client(sock_dir, "a")
    b = make_client(sock_dir, "b")
    a.send(proto.REQ_LOCK)
    assert a.recv(5).type == proto.LOCK_OK
    t0 = time.monotonic()
    b.send(proto.REQ_LOCK)
    # TQ=1s (sched fixture): DROP_LOCK should arrive in ~1s.
    msg = a.recv(5)
    dt = time.monotonic() - t0
    assert msg.type == proto.DROP_LOCK
    assert 0.3 < dt < 4.5, f"preemption after {dt:.2f}s with TQ=1"
    a.send(proto.LOCK_RELEASED)
    assert b.recv(5).type == proto.LOCK_OK
    a.close()
    b.close()


def test_solo_holder_not_preempted(sched, sock_dir):
    """nvshare-amd solo fast path: no DROP_LOCK without waiters."""
    a = make_client(sock_dir, "a")
    a.send(proto.REQ_LOCK)
    assert a.recv(5).type == proto.LOCK_OK
    with pytest.raises(Exception):
        a.recv(2.5)  # TQ=1: reference would preempt; we must not
    a.close()


def test_waiter_arrival_preempts_overdue_holder(sched, sock_dir):
    """A holder past its quantum is preempted soon after a waiter shows."""
    a = make_client(sock_dir, "a")
    b = make_client(sock_dir, "b")
    a.send(proto.REQ_LOCK)
    assert a.recv(5).type == proto.LOCK_OK
    time.sleep(1.5)  # exceed TQ=1 while solo
    t0 = time.monotonic()
    b.send(proto.REQ_LOCK)
    msg = a.recv(5)
    dt = time.monotonic() - t0
    assert msg.type == proto.DROP_LOCK
    assert dt < 2.5, f"overdue holder preempted after {dt:.2f}s"
    a.close()
    b.close()


def test_set_tq(sched, sock_dir):
    ctl.set_tq(7, sock_dir)
    st = ctl.status(sock_dir)
    assert st.tq_seconds == 7


def test_sched_off_on_broadcast(sched, sock_dir):
    a = make_client(sock_dir, "a")
    ctl.set_scheduling(False, sock_dir)
    assert a.recv(5).type == proto.SCHED_OFF
    assert not ctl.status(sock_dir).scheduling_on
    ctl.set_scheduling(True, sock_dir)
    assert a.recv(5).type == proto.SCHED_ON
    assert ctl.status(sock_dir).scheduling_on
    a.close()


def test_sched_off_flushes_queue_and_lock(sched, sock_dir):
    a = make_client(sock_dir, "a")
    b = make_client(sock_dir, "b")
    a.send(proto.REQ_LOCK)
    assert a.recv(5).type == proto.LOCK_OK
    b.send(proto.REQ_LOCK)
    ctl.set_scheduling(False, sock_dir)
    assert ctl.status(sock_dir).queued == 0
    ctl.set_scheduling(True, sock_dir)
    # Both clients free-ran; re-request works.
    for c in (a, b):
        while True:  # drain broadcasts
            m = c.recv(5)
            if m.type in (proto.SCHED_ON,):
                break
    a.send(proto.REQ_LOCK)
    assert a.recv(5).type == proto.LOCK_OK
    a.close()
    b.close()


def test_eviction_on_disconnect(sched, sock_dir):
    a = make_client(sock_dir, "a")
    b = make_client(sock_dir, "b")
    a.send(proto.REQ_LOCK)
    assert a.recv(5).type == proto.LOCK_OK
    b.send(proto.REQ_LOCK)
    a.close()  # holder dies -> lock must pass to b
    assert b.recv(5).type == proto.LOCK_OK
    st = ctl.status(sock_dir)
    assert st.clients == 1
    b.close()


def test_status_counts(sched, sock_dir):
    st = ctl.status(sock_dir)
    assert st.clients == 0 and st.queued == 0
    a = make_client(sock_dir, "a")
    st = ctl.status(sock_dir)
    assert st.clients == 1
    a.close()
    time.sleep(0.2)
    st = ctl.status(sock_dir)
    assert st.clients == 0


def test_set_tq_extends_current_quantum(sched, sock_dir):
    """Raising TQ mid-quantum defers the pending preemption."""
    a = make_client(sock_dir, "a")
    b = make_client(sock_dir, "b")
    a.send(proto.REQ_LOCK)
    assert a.recv(5).type == proto.LOCK_OK
    b.send(proto.REQ_LOCK)          # arms the TQ=1 preemption
    ctl.set_tq(30, sock_dir)        # extend before it fires
    with pytest.raises(Exception):
        a.recv(2.5)                 # no DROP_LOCK under the new TQ
    ctl.set_tq(1, sock_dir)         # shrink back: holder is overdue
    assert a.recv(5).type == proto.DROP_LOCK
    a.send(proto.LOCK_RELEASED)
    assert b.recv(5).type == proto.LOCK_OK
    a.close()
    b.close()


def test_mem_update_from_unregistered_ignored(sched, sock_dir):
    s = proto.Client(sock_dir=sock_dir, pod_name="raw").connect()
    proto.send_msg(s.sock, proto.Message(proto.MEM_UPDATE, data="999"))
    time.sleep(0.2)
    assert ctl.status(sock_dir).tracked_mib == 0
    s.close()


def test_mem_update_reported_in_status(sched, sock_dir):
    a = make_client(sock_dir, "a")
    b = make_client(sock_dir, "b")
    a.send(proto.MEM_UPDATE, "2048")
    b.send(proto.MEM_UPDATE, "1024")
    time.sleep(0.2)
    st = ctl.status(sock_dir)
    assert st.tracked_mib == 3072
    a.send(proto.MEM_UPDATE, "512")
    time.sleep(0.2)
    assert ctl.status(sock_dir).tracked_mib == 1536
    a.close()
    time.sleep(0.2)
    assert ctl.status(sock_dir).tracked_mib == 1024
    b.close()


def test_pressure_flag_in_lock_ok(sched, sock_dir):
    """LOCK_OK carries the scheduler's whole-node pressure verdict:
    p=1 when the clients' combined tracked sets exceed the reported
    device capacity, p=0 when they fit, no data when no client has
    reported a capacity (reference-era clients)."""
    a = make_client(sock_dir, "a")
    b = make_client(sock_dir, "b")

    # No capacity reported yet: grant carries no verdict.
    a.send(proto.REQ_LOCK)
    m = a.recv(5)
    assert m.type == proto.LOCK_OK
    assert m.data == ""
    a.send(proto.LOCK_RELEASED)

    # Fits: 1024 + 512 < 4096.
    a.send(proto.MEM_UPDATE, "1024,4096")
    b.send(proto.MEM_UPDATE, "512,4096")
    time.sleep(0.2)
    a.send(proto.REQ_LOCK)
    m = a.recv(5)
    assert m.type == proto.LOCK_OK
    assert m.data == "p=0"
    a.send(proto.LOCK_RELEASED)

    # Oversubscribed: 3000 + 2000 > 4096.
    a.send(proto.MEM_UPDATE, "3000,4096")
    b.send(proto.MEM_UPDATE, "2000,4096")
    time.sleep(0.2)
    b.send(proto.REQ_LOCK)
    m = b.recv(5)
    assert m.type == proto.LOCK_OK
    assert m.data == "p=1"
    a.close()
    b.close()


def test_stale_lock_released_ignored(sched, sock_dir):
    a = make_client(sock_dir, "a")
    b = make_client(sock_dir, "b")
    b.send(proto.LOCK_RELEASED)  # b never held the lock
    a.send(proto.REQ_LOCK)
    assert a.recv(5).type == proto.LOCK_OK
    a.close()
    b.close()


def test_multi_gpu_independent_locks(sched, sock_dir):
    """Clients on different GPUs hold their locks CONCURRENTLY; the
    reference was single-GPU (README.md:97) — nvshare-amd arbitrates
    each of a node's 8 MI355Xs independently."""
    a = make_client(sock_dir, "a", gpu=0)
    b = make_client(sock_dir, "b", gpu=1)
    a.send(proto.REQ_LOCK)
    b.send(proto.REQ_LOCK)
    assert a.recv(5).type == proto.LOCK_OK
    assert b.recv(5).type == proto.LOCK_OK  # no waiting on gpu0's lock
    # Neither gets preempted (each is solo on its GPU, TQ=1).
    with pytest.raises(Exception):
        a.recv(2.0)
    a.close()
    b.close()


def test_multi_gpu_preemption_is_per_gpu(sched, sock_dir):
    """A waiter on gpu1 must not preempt the holder on gpu0."""
    a = make_client(sock_dir, "a", gpu=0)
    b = make_client(sock_dir, "b", gpu=1)
    c = make_client(sock_dir, "c", gpu=1)
    a.send(proto.REQ_LOCK)
    assert a.recv(5).type == proto.LOCK_OK
    b.send(proto.REQ_LOCK)
    assert b.recv(5).type == proto.LOCK_OK
    c.send(proto.REQ_LOCK)  # waits behind b on gpu1
    # b gets DROP_LOCK after ~TQ; a (solo on gpu0) must not.
    assert b.recv(5).type == proto.DROP_LOCK
    b.send(proto.LOCK_RELEASED)
    assert c.recv(5).type == proto.LOCK_OK
    with pytest.raises(Exception):
        a.recv(1.0)
    for x in (a, b, c):
        x.close()


def test_env_tq_startup(artifacts, sock_dir):
    from nvshare_amd.scheduler import SchedulerDaemon

    with SchedulerDaemon(sock_dir=sock_dir, tq=42):
        assert ctl.status(sock_dir).tq_seconds == 42


def test_env_sched_off_startup(artifacts, sock_dir):
    from nvshare_amd.scheduler import SchedulerDaemon

    with SchedulerDaemon(sock_dir=sock_dir, sched_off=True):
        assert not ctl.status(sock_dir).scheduling_on
        c = proto.Client(sock_dir=sock_dir, pod_name="x")
        c.connect()
        reply = c.register()
        assert reply.type == proto.SCHED_OFF
        c.close()


def test_per_gpu_tq_override(sched, sock_dir):
    """SET_TQ "gpuN:tq" overrides one GPU's quantum only."""
    ctl.set_tq(30, sock_dir)           # global: long
    ctl.set_tq(1, sock_dir, gpu=1)     # gpu1: short
    a0 = make_client(sock_dir, "a0", gpu=0)
    b0 = make_client(sock_dir, "b0", gpu=0)
    a1 = make_client(sock_dir, "a1", gpu=1)
    b1 = make_client(sock_dir, "b1", gpu=1)
    a0.send(proto.REQ_LOCK)
    assert a0.recv(5).type == proto.LOCK_OK
    a1.send(proto.REQ_LOCK)
    assert a1.recv(5).type == proto.LOCK_OK
    b0.send(proto.REQ_LOCK)
    b1.send(proto.REQ_LOCK)
    # gpu1's holder preempts after ~1 s; gpu0's not within 3 s.
    assert a1.recv(5).type == proto.DROP_LOCK
    with pytest.raises(Exception):
        a0.recv(2.0)
    for c in (a0, b0, a1, b1):
        c.close()


def test_round_robin_rotation(sched, sock_dir):
    """Three clients that re-request after each release rotate in FCFS
    order across multiple quanta (A, B, C, A, B, ...)."""
    clients = {n: make_client(sock_dir, n) for n in "abc"}
    order = []
    for n in "abc":
        clients[n].send(proto.REQ_LOCK)
        time.sleep(0.05)
    for _ in range(6):
        granted = None
        for n, c in clients.items():
            try:
                m = c.recv(0.1)
            except Exception:
                continue
            if m.type == proto.LOCK_OK:
                granted = n
                break
        if granted is None:
            # wait for the next grant on any socket
            for n, c in clients.items():
                try:
                    m = c.recv(3)
                except Exception:
                    continue
                if m.type == proto.LOCK_OK:
                    granted = n
                    break
        assert granted is not None
        order.append(granted)
        clients[granted].send(proto.LOCK_RELEASED)
        clients[granted].send(proto.REQ_LOCK)  # back of the queue
    # Strict FCFS: the rotation must cycle a,b,c,a,b,c.
    assert order == list("abcabc"), order
    for c in clients.values():
        c.close()
