"""Python mirror of the nvshare-amd wire protocol (src/proto.h).

537-byte packed message over a Unix stream socket, layout-compatible
with the reference protocol (reference src/comm.h:59-80):
    uint8   type
    char    pod_name[254]
    char    pod_namespace[254]
    uint64  id (little-endian)
    char    data[20]
"""

from __future__ import annotations

import os
import socket
import struct
from dataclasses import dataclass, field

MSG_SIZE = 537
_FMT = "<B254s254sQ20s"
assert struct.calcsize(_FMT) == MSG_SIZE

REGISTER = 1
SCHED_ON = 2
SCHED_OFF = 3
REQ_LOCK = 4
LOCK_OK = 5
DROP_LOCK = 6
LOCK_RELEASED = 7
SET_TQ = 8
STATUS_REQ = 9
STATUS = 10
MEM_UPDATE = 11

TYPE_NAMES = {
    REGISTER: "REGISTER",
    SCHED_ON: "SCHED_ON",
    SCHED_OFF: "SCHED_OFF",
    REQ_LOCK: "REQ_LOCK",
    LOCK_OK: "LOCK_OK",
    DROP_LOCK: "DROP_LOCK",
    LOCK_RELEASED: "LOCK_RELEASED",
    SET_TQ: "SET_TQ",
    STATUS_REQ: "STATUS_REQ",
    STATUS: "STATUS",
    MEM_UPDATE: "MEM_UPDATE",
}

SOCK_DIR_DEFAULT = "/var/run/nvshare/"
SOCK_NAME = "scheduler.sock"


def scheduler_path(sock_dir: str | None = None) -> str:
    d = sock_dir or os.environ.get("NVSHARE_SOCK_DIR") or SOCK_DIR_DEFAULT
    return os.path.join(d, SOCK_NAME)


@dataclass
class Message:
    type: int
    pod_name: str = ""
    pod_namespace: str = ""
    id: int = 0
    data: str = ""

    def pack(self) -> bytes:
        return struct.pack(
            _FMT,
            self.type,
            self.pod_name.encode()[:253],
            self.pod_namespace.encode()[:253],
            self.id,
            self.data.encode()[:19],
        )

    @classmethod
    def unpack(cls, raw: bytes) -> "Message":
        t, name, ns, mid, data = struct.unpack(_FMT, raw)
        z = lambda b: b.split(b"\0", 1)[0].decode(errors="replace")
        return cls(t, z(name), z(ns), mid, z(data))

    @property
    def type_name(self) -> str:
        return TYPE_NAMES.get(self.type, f"UNKNOWN({self.type})")


def recv_msg(sock: socket.socket, timeout: float | None = None) -> Message:
    sock.settimeout(timeout)
    buf = b""
    while len(buf) < MSG_SIZE:
        chunk = sock.recv(MSG_SIZE - len(buf))
        if not chunk:
            raise ConnectionError("scheduler closed the connection")
        buf += chunk
    return Message.unpack(buf)


def send_msg(sock: socket.socket, msg: Message) -> None:
    sock.sendall(msg.pack())


@dataclass
class Client:
    """A scriptable protocol client (used by tests and tooling)."""

    sock_dir: str | None = None
    pod_name: str = "pyclient"
    pod_namespace: str = ""
    sock: socket.socket | None = None
    client_id: int = 0
    scheduling_on: bool = field(default=True)
    gpu: int = 0

    def connect(self, timeout: float = 10.0) -> "Client":
        self.sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self.sock.settimeout(timeout)
        self.sock.connect(scheduler_path(self.sock_dir))
        return self

    def register(self, timeout: float = 10.0) -> Message:
        assert self.sock is not None
        send_msg(self.sock, Message(REGISTER, self.pod_name,
                                    self.pod_namespace,
                                    data=f"gpu{self.gpu}"))
        reply = recv_msg(self.sock, timeout)
        if reply.type not in (SCHED_ON, SCHED_OFF):
            raise RuntimeError(f"unexpected handshake: {reply.type_name}")
        self.scheduling_on = reply.type == SCHED_ON
        self.client_id = int(reply.data, 16) if reply.data else 0
        return reply

    def send(self, msg_type: int, data: str = "") -> None:
        assert self.sock is not None
        send_msg(self.sock, Message(msg_type, self.pod_name,
                                    self.pod_namespace, self.client_id,
                                    data))

    def recv(self, timeout: float | None = 10.0) -> Message:
        assert self.sock is not None
        return recv_msg(self.sock, timeout)

    def close(self) -> None:
        if self.sock is not None:
            self.sock.close()
            self.sock = None

    def __enter__(self) -> "Client":
        return self.connect()

    def __exit__(self, *exc) -> None:
        self.close()
