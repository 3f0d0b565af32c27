"""Wire-protocol unit tests (no GPU).

The message layout must stay byte-compatible with the reference
protocol (reference src/comm.h:70-80: 537-byte packed struct).
"""

import struct

from nvshare_amd import proto


def test_message_size():
    assert proto.MSG_SIZE == 537
    assert len(proto.Message(proto.REGISTER).pack()) == 537


def test_pack_layout():
    m = proto.Message(proto.SET_TQ, "pod", "ns", 0x1122334455667788, "60")
    raw = m.pack()
    assert raw[0] == proto.SET_TQ
    assert raw[1:4] == b"pod"
    assert raw[255:257] == b"ns"
    # id at offset 1+254+254 = 509, little-endian u64
    assert struct.unpack_from("<Q", raw, 509)[0] == 0x1122334455667788
    assert raw[517:519] == b"60"


def test_roundtrip():
    m = proto.Message(proto.LOCK_OK, "name", "namespace", 42, "data")
    m2 = proto.Message.unpack(m.pack())
    assert m2 == m


def test_truncation():
    m = proto.Message(proto.REGISTER, "x" * 500, "y" * 500, 0, "z" * 50)
    m2 = proto.Message.unpack(m.pack())
    assert len(m2.pod_name) == 253
    assert len(m2.pod_namespace) == 253
    assert len(m2.data) == 19


def test_type_names():
    assert proto.TYPE_NAMES[1] == "REGISTER"
    assert proto.TYPE_NAMES[8] == "SET_TQ"


def test_reference_shaped_client_compat(sched, sock_dir):
    """A reference-shaped client (REGISTER with empty data, no
    extension types) gets the full lock lifecycle on gpu0."""
    import socket as socklib

    from nvshare_amd import proto

    s = socklib.socket(socklib.AF_UNIX, socklib.SOCK_STREAM)
    s.settimeout(5)
    s.connect(proto.scheduler_path(sock_dir))
    # exactly what the reference sends: type=REGISTER, empty data
    proto.send_msg(s, proto.Message(proto.REGISTER, "refpod", "refns"))
    reply = proto.recv_msg(s, 5)
    assert reply.type == proto.SCHED_ON
    assert len(reply.data) == 16  # hex client id
    proto.send_msg(s, proto.Message(proto.REQ_LOCK))
    assert proto.recv_msg(s, 5).type == proto.LOCK_OK
    proto.send_msg(s, proto.Message(proto.LOCK_RELEASED))
    proto.send_msg(s, proto.Message(proto.REQ_LOCK))
    assert proto.recv_msg(s, 5).type == proto.LOCK_OK
    s.close()
