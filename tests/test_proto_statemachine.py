"""Randomized protocol state-machine test (seeded, deterministic).

Drives the daemon with random interleavings of the full client
vocabulary and checks the scheduler's core invariants after every
step — property-style depth beyond the scripted lifecycle tests:

  I1  at most one client believes it holds a GPU's lock
  I2  the daemon keeps answering STATUS (no wedge)
  I3  a granted lock is always eventually revocable: after every
      client disconnects, a fresh client gets the lock immediately
"""

from __future__ import annotations

import random
import time

from nvshare_amd import ctl, proto


class Peer:
    def __init__(self, sock_dir, name):
        self.c = proto.Client(sock_dir=sock_dir, pod_name=name)
        self.c.connect()
        self.c.register()
        self.holding = False
        self.requested = False
        self.alive = True

    def drain(self):
        """Consume any pending messages, tracking believed lock state."""
        while True:
            try:
                m = self.c.recv(0.02)
            except Exception:
                return
            if m.type == proto.LOCK_OK:
                self.holding = True
                self.requested = False
            elif m.type == proto.DROP_LOCK:
                if self.holding:
                    self.holding = False
                    self.c.send(proto.LOCK_RELEASED)
            elif m.type in (proto.SCHED_ON, proto.SCHED_OFF):
                self.holding = False
                self.requested = False


def test_random_interleavings(sched, sock_dir):
    rng = random.Random(20260914)
    peers = [Peer(sock_dir, f"p{i}") for i in range(4)]
    spawned = len(peers)

    for step in range(300):
        p = rng.choice(peers)
        op = rng.randrange(7)
        if not p.alive:
            continue
        if op == 0 and not p.requested and not p.holding:
            p.c.send(proto.REQ_LOCK)
            p.requested = True
        elif op == 1 and p.holding:
            p.c.send(proto.LOCK_RELEASED)
            p.holding = False
        elif op == 2:
            p.c.send(proto.MEM_UPDATE,
                     f"{rng.randrange(0, 5000)},4096")
        elif op == 3 and rng.random() < 0.1:
            p.c.close()
            p.alive = False
        elif op == 4 and rng.random() < 0.2:
            ctl.set_tq(rng.choice([1, 2, 30]), sock_dir)
        elif op == 5:
            pass  # idle tick
        for q in peers:
            if q.alive:
                q.drain()

        # I1: at most one believed holder.
        holders = [q for q in peers if q.alive and q.holding]
        assert len(holders) <= 1, [q.c.pod_name for q in holders]

        # Keep the population alive.
        if sum(q.alive for q in peers) < 2:
            peers.append(Peer(sock_dir, f"p{spawned}"))
            spawned += 1

        # I2 (sampled): daemon still answers.
        if step % 50 == 0:
            st = ctl.status(sock_dir, timeout=5)
            assert st.clients >= 1

    # Teardown: release/close everyone.
    for q in peers:
        if q.alive:
            if q.holding:
                q.c.send(proto.LOCK_RELEASED)
            q.c.close()
    time.sleep(0.3)

    # I3: a fresh client gets the lock immediately.
    z = Peer(sock_dir, "final")
    z.c.send(proto.REQ_LOCK)
    m = z.c.recv(5)
    assert m.type == proto.LOCK_OK
    z.c.close()

    st = ctl.status(sock_dir, timeout=5)
    assert st.clients == 0
