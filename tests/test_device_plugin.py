"""K8s device plugin tests with a fake kubelet (no cluster, no GPU).

Drives the real gRPC server over a Unix socket: registration,
ListAndWatch device advertisement, and Allocate-time env/mount/device
injection (the contract the reference's Go plugin implements,
server.go:219-277).
"""

from __future__ import annotations

import tempfile
import threading
import time
from concurrent import futures

import grpc
import pytest

from nvshare_amd.k8s import pb
from nvshare_amd.k8s.device_plugin import (
    DevicePlugin, PluginConfig, device_id, parse_device_id,
)


@pytest.fixture
def kubelet_dir():
    with tempfile.TemporaryDirectory(prefix="kubelet-",
                                     dir="/tmp") as d:
        yield d


@pytest.fixture
def plugin(kubelet_dir):
    cfg = PluginConfig(virtual_devices=4, kubelet_dir=kubelet_dir,
                       gpus=[0], lib_dir="/usr/lib/nvshare",
                       sock_dir="/var/run/nvshare")
    p = DevicePlugin(cfg)
    p.start()
    yield p
    p.stop()


def plugin_channel(plugin):
    return grpc.insecure_channel(f"unix://{plugin.endpoint}")


def test_device_id_roundtrip():
    assert parse_device_id(device_id(3, 7)) == (3, 7)
    with pytest.raises(ValueError):
        parse_device_id("bogus")


def test_get_options(plugin):
    with plugin_channel(plugin) as chan:
        rpc = chan.unary_unary(
            f"/{pb.DEVICE_PLUGIN_SERVICE}/GetDevicePluginOptions",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=pb.DevicePluginOptions.FromString)
        opts = rpc(pb.Empty(), timeout=5)
    assert opts.pre_start_required is False


def test_list_and_watch_devices(plugin):
    with plugin_channel(plugin) as chan:
        rpc = chan.unary_stream(
            f"/{pb.DEVICE_PLUGIN_SERVICE}/ListAndWatch",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=pb.ListAndWatchResponse.FromString)
        stream = rpc(pb.Empty(), timeout=10)
        first = next(stream)
        stream.cancel()
    ids = [d.ID for d in first.devices]
    assert len(ids) == 4
    assert ids[0] == "gpu0__0"
    assert all(d.health == "Healthy" for d in first.devices)


def test_allocate_injects_env_and_mounts(plugin):
    req = pb.AllocateRequest()
    creq = req.container_requests.add()
    creq.devicesIDs.append("gpu0__2")
    with plugin_channel(plugin) as chan:
        rpc = chan.unary_unary(
            f"/{pb.DEVICE_PLUGIN_SERVICE}/Allocate",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=pb.AllocateResponse.FromString)
        resp = rpc(req, timeout=5)
    assert len(resp.container_responses) == 1
    c = resp.container_responses[0]
    assert c.envs["LD_PRELOAD"] == "/usr/lib/nvshare/libnvshare.so"
    assert c.envs["HSA_XNACK"] == "1"
    assert c.envs["NVSHARE_SOCK_DIR"] == "/var/run/nvshare/"
    # Containers see their GPU as HIP device 0; the node-level index
    # routes the client to the right per-GPU arbitration queue.
    assert c.envs["NVSHARE_GPU"] == "0"
    paths = [d.container_path for d in c.devices]
    assert "/dev/kfd" in paths
    mounts = {m.container_path: m for m in c.mounts}
    assert "/usr/lib/nvshare/libnvshare.so" in mounts
    assert mounts["/usr/lib/nvshare/libnvshare.so"].read_only


def test_allocate_malformed_id(plugin):
    req = pb.AllocateRequest()
    creq = req.container_requests.add()
    creq.devicesIDs.append("not-a-device")
    with plugin_channel(plugin) as chan:
        rpc = chan.unary_unary(
            f"/{pb.DEVICE_PLUGIN_SERVICE}/Allocate",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=pb.AllocateResponse.FromString)
        with pytest.raises(grpc.RpcError):
            rpc(req, timeout=5)


def test_register_with_fake_kubelet(plugin, kubelet_dir):
    """A fake kubelet Registration service records the plugin's
    RegisterRequest."""
    got = {}
    ev = threading.Event()

    def register(req_bytes, ctx):
        req = pb.RegisterRequest.FromString(req_bytes)
        got["version"] = req.version
        got["endpoint"] = req.endpoint
        got["resource"] = req.resource_name
        ev.set()
        return pb.Empty()

    handler = grpc.method_handlers_generic_handler(
        pb.REGISTRATION_SERVICE,
        {"Register": grpc.unary_unary_rpc_method_handler(
            register,
            response_serializer=lambda m: m.SerializeToString())})
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
    server.add_generic_rpc_handlers((handler,))
    server.add_insecure_port(f"unix://{kubelet_dir}/kubelet.sock")
    server.start()
    try:
        plugin.register(timeout=5)
        assert ev.wait(5)
        assert got["version"] == "v1beta1"
        assert got["endpoint"] == "nvshare-amd.sock"
        assert got["resource"] == "nvshare.com/gpu"
    finally:
        server.stop(grace=1)


def test_restart_on_kubelet_socket_recreation(kubelet_dir):
    """run_forever re-registers when the kubelet socket is recreated
    (reference main.go:132-177 inotify restart loop)."""
    registrations = []
    ev = threading.Event()

    def register(req_bytes, ctx):
        registrations.append(pb.RegisterRequest.FromString(req_bytes))
        ev.set()
        return pb.Empty()

    handler = grpc.method_handlers_generic_handler(
        pb.REGISTRATION_SERVICE,
        {"Register": grpc.unary_unary_rpc_method_handler(
            register,
            response_serializer=lambda m: m.SerializeToString())})

    def make_kubelet():
        server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
        server.add_generic_rpc_handlers((handler,))
        server.add_insecure_port(
            f"unix://{kubelet_dir}/kubelet.sock")
        server.start()
        return server

    k1 = make_kubelet()
    cfg = PluginConfig(virtual_devices=2, kubelet_dir=kubelet_dir,
                       gpus=[0])
    p = DevicePlugin(cfg)
    t = threading.Thread(target=p.run_forever, daemon=True)
    t.start()
    try:
        assert ev.wait(10), "first registration missing"
        ev.clear()
        n_first = len(registrations)
        # Let the plugin snapshot the current socket inode before we
        # swap it (it stats right after register returns).
        time.sleep(1.0)
        # Recreate the kubelet socket (inode changes).
        k1.stop(grace=0)
        import contextlib
        import os as _os
        with contextlib.suppress(FileNotFoundError):
            _os.unlink(f"{kubelet_dir}/kubelet.sock")
        k2 = make_kubelet()
        assert ev.wait(15), "no re-registration after recreation"
        assert len(registrations) > n_first
        k2.stop(grace=0)
    finally:
        p.stop()


def test_reregister_after_kubelet_outage(kubelet_dir):
    """Socket deleted with no kubelet for a while: the plugin keeps
    retrying and re-registers when the kubelet comes back (reference
    main.go restart loop survives kubelet downtime)."""
    registrations = []
    ev = threading.Event()

    def register(req_bytes, ctx):
        registrations.append(pb.RegisterRequest.FromString(req_bytes))
        ev.set()
        return pb.Empty()

    handler = grpc.method_handlers_generic_handler(
        pb.REGISTRATION_SERVICE,
        {"Register": grpc.unary_unary_rpc_method_handler(
            register,
            response_serializer=lambda m: m.SerializeToString())})

    def make_kubelet():
        server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
        server.add_generic_rpc_handlers((handler,))
        server.add_insecure_port(f"unix://{kubelet_dir}/kubelet.sock")
        server.start()
        return server

    k1 = make_kubelet()
    cfg = PluginConfig(virtual_devices=2, kubelet_dir=kubelet_dir,
                       gpus=[0])
    p = DevicePlugin(cfg)
    t = threading.Thread(target=p.run_forever, daemon=True)
    t.start()
    try:
        assert ev.wait(10), "first registration missing"
        ev.clear()
        n_first = len(registrations)
        time.sleep(1.0)
        # Kubelet dies and its socket disappears; nothing replaces it
        # for several seconds.
        k1.stop(grace=0)
        import contextlib
        import os as _os
        with contextlib.suppress(FileNotFoundError):
            _os.unlink(f"{kubelet_dir}/kubelet.sock")
        time.sleep(5.0)
        assert not ev.is_set(), "spurious registration with no kubelet"
        # Kubelet returns: the plugin must re-register on its own.
        k2 = make_kubelet()
        assert ev.wait(20), "no re-registration after kubelet outage"
        assert len(registrations) > n_first
        k2.stop(grace=0)
    finally:
        p.stop()


def test_preferred_allocation(plugin):
    req = pb.PreferredAllocationRequest()
    creq = req.container_requests.add()
    creq.available_deviceIDs.extend(["gpu0__0", "gpu0__1", "gpu0__2"])
    creq.allocation_size = 2
    with plugin_channel(plugin) as chan:
        rpc = chan.unary_unary(
            f"/{pb.DEVICE_PLUGIN_SERVICE}/GetPreferredAllocation",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=(
                pb.PreferredAllocationResponse.FromString))
        resp = rpc(req, timeout=5)
    assert list(resp.container_responses[0].deviceIDs) == \
        ["gpu0__0", "gpu0__1"]


def test_allocate_env_drives_client_to_gpu_queue(plugin, sock_dir,
                                                 artifacts):
    """End-to-end bridge: the EXACT env the plugin injects (plus the
    CPU stub lib path) makes a client register on the allocated GPU's
    queue of a live scheduler."""
    import os
    import subprocess

    from nvshare_amd import ctl
    from nvshare_amd.scheduler import SchedulerDaemon

    req = pb.AllocateRequest()
    creq = req.container_requests.add()
    creq.devicesIDs.append("gpu3__1")
    with plugin_channel(plugin) as chan:
        rpc = chan.unary_unary(
            f"/{pb.DEVICE_PLUGIN_SERVICE}/Allocate",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=pb.AllocateResponse.FromString)
        resp = rpc(req, timeout=5)
    c = resp.container_responses[0]
    assert c.envs["NVSHARE_GPU"] == "3"

    with SchedulerDaemon(sock_dir=sock_dir, tq=5, debug=True) as sched:
        env = dict(os.environ)
        env.update(dict(c.envs))
        # Container-equivalent substitutions for the CPU harness: the
        # real mounts map libnvshare.so + the socket dir into the pod.
        env["LD_PRELOAD"] = str(artifacts.libnvshare)
        env["NVSHARE_SOCK_DIR"] = sock_dir
        env["LD_LIBRARY_PATH"] = str(artifacts.stub_dir)
        env["NVSTUB_TOTAL_MIB"] = "1024"
        env["NVSHARE_RESERVE_MIB"] = "64"
        env["NVSHARE_DEBUG"] = "1"
        r = subprocess.run(
            [str(artifacts.hipclient), "--allocs", "1", "--alloc-mib",
             "8", "--iters", "5"],
            env=env, capture_output=True, text=True, timeout=60)
        assert r.returncode == 0, (r.stdout, r.stderr[-2000:])
        assert "PASS" in r.stdout
        log = sched.log_text()
        assert "on gpu3" in log, log[-2000:]


def test_gate_only_client_mode(kubelet_dir):
    """NVSHARE_CLIENT_MODE=gate-only deployments inject
    NVSHARE_DISABLE_UM=1: real-VRAM clients, scheduler arbitration and
    cap only (zero-overhead sharing for fits-in-HBM fleets,
    profiles/RESULTS.md §18)."""
    cfg = PluginConfig(virtual_devices=2, kubelet_dir=kubelet_dir,
                       gpus=[0], client_mode="gate-only")
    p = DevicePlugin(cfg)
    p.start()
    try:
        req = pb.AllocateRequest()
        creq = req.container_requests.add()
        creq.devicesIDs.append("gpu0__0")
        with plugin_channel(p) as chan:
            rpc = chan.unary_unary(
                f"/{pb.DEVICE_PLUGIN_SERVICE}/Allocate",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=pb.AllocateResponse.FromString)
            resp = rpc(req, timeout=5)
        c = resp.container_responses[0]
        assert c.envs["NVSHARE_DISABLE_UM"] == "1"
    finally:
        p.stop()
