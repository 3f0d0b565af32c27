"""Control-plane client: configure/query a running scheduler.

Python counterpart of nvsharectl (src/ctl.c); used by tests, the bench
harness and ops tooling.
"""

from __future__ import annotations

import socket
from dataclasses import dataclass

from nvshare_amd import proto


@dataclass
class SchedulerStatus:
    scheduling_on: bool
    tq_seconds: int
    clients: int
    queued: int
    tracked_mib: int = 0


def _one_shot(msg_type: int, data: str = "",
              sock_dir: str | None = None,
              want_reply: bool = False,
              timeout: float = 10.0) -> proto.Message | None:
    s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    s.settimeout(timeout)
    try:
        s.connect(proto.scheduler_path(sock_dir))
        proto.send_msg(s, proto.Message(msg_type, data=data))
        if want_reply:
            return proto.recv_msg(s, timeout)
        return None
    finally:
        s.close()


def set_tq(seconds: int, sock_dir: str | None = None,
           gpu: int | None = None) -> None:
    if not 1 <= seconds <= 86400:
        raise ValueError("TQ must be in [1, 86400] seconds")
    data = f"gpu{gpu}:{seconds}" if gpu is not None else str(seconds)
    _one_shot(proto.SET_TQ, data, sock_dir)


def set_scheduling(on: bool, sock_dir: str | None = None) -> None:
    _one_shot(proto.SCHED_ON if on else proto.SCHED_OFF, "", sock_dir)


def status(sock_dir: str | None = None,
           timeout: float = 10.0) -> SchedulerStatus:
    reply = _one_shot(proto.STATUS_REQ, "", sock_dir, want_reply=True,
                      timeout=timeout)
    assert reply is not None and reply.type == proto.STATUS, reply
    parts = [int(x) for x in reply.data.split(",")]
    on, tq, ncl, qlen = parts[:4]
    mib = parts[4] if len(parts) > 4 else 0
    return SchedulerStatus(bool(on), tq, ncl, qlen, mib)
