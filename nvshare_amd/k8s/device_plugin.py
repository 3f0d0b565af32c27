"""nvshare-amd Kubernetes device plugin (kubelet v1beta1, grpcio).

Advertises NVSHARE_VIRTUAL_DEVICES virtual `nvshare.com/gpu` devices
per physical MI355X and, on Allocate, injects into the requesting
container everything an nvshare client needs:
  - LD_PRELOAD=<lib dir>/libnvshare.so  (interposer)
  - HSA_XNACK=1                         (gfx950 demand paging)
  - NVSHARE_SOCK_DIR                    (scheduler socket)
  - /dev/kfd + the GPU's /dev/dri nodes (AMD device exposure — the
    NVIDIA reference used NVIDIA_VISIBLE_DEVICES instead,
    server.go:233-241; AMD GPUs are plain device nodes)
  - read-only mounts of libnvshare.so and the scheduler socket dir
    (reference server.go:243-272)

Behavioral parity reference: kubernetes/device-plugin/*.go (Go, gRPC).
This is a from-scratch Python implementation because the build
environment has no Go toolchain; the wire API is identical.
"""

from __future__ import annotations

import glob
import logging
import os
import re
import threading
import time
from concurrent import futures
from dataclasses import dataclass, field

import grpc

from nvshare_amd.k8s import pb

log = logging.getLogger("nvshare.device-plugin")

KUBELET_DIR = "/var/lib/kubelet/device-plugins"
KUBELET_SOCK = "kubelet.sock"
PLUGIN_SOCK = "nvshare-amd.sock"
RESOURCE_NAME = "nvshare.com/gpu"
HEALTHY = "Healthy"

LIB_DIR_DEFAULT = "/usr/lib/nvshare"
SOCK_DIR_DEFAULT = "/var/run/nvshare"


def discover_gpus() -> list[int]:
    """Physical AMD GPU indices via /dev/dri render nodes."""
    nodes = sorted(glob.glob("/dev/dri/renderD*"))
    return list(range(len(nodes)))


def render_nodes() -> list[str]:
    return sorted(glob.glob("/dev/dri/renderD*"))


def device_id(gpu: int, ordinal: int) -> str:
    # reference devices.go:14-20 used "<UUID>__<ordinal>"; AMD GPUs
    # have no UUID env protocol, so "<gpuN>__<ordinal>".
    return f"gpu{gpu}__{ordinal}"


_ID_RE = re.compile(r"^gpu(\d+)__(\d+)$")


def parse_device_id(dev_id: str) -> tuple[int, int]:
    m = _ID_RE.match(dev_id)
    if not m:
        raise ValueError(f"malformed device id {dev_id!r}")
    return int(m.group(1)), int(m.group(2))


@dataclass
class PluginConfig:
    virtual_devices: int = int(os.environ.get(
        "NVSHARE_VIRTUAL_DEVICES", "10"))
    lib_dir: str = os.environ.get("NVSHARE_LIB_DIR", LIB_DIR_DEFAULT)
    sock_dir: str = os.environ.get("NVSHARE_SOCK_DIR", SOCK_DIR_DEFAULT)
    kubelet_dir: str = os.environ.get("NVSHARE_KUBELET_DIR", KUBELET_DIR)
    resource_name: str = os.environ.get("NVSHARE_RESOURCE_NAME",
                                        RESOURCE_NAME)
    gpus: list[int] = field(default_factory=discover_gpus)
    debug_clients: bool = bool(os.environ.get("NVSHARE_CLIENT_DEBUG"))
    # "managed": hipMalloc -> hipMallocManaged (oversubscribable, the
    # reference behavior; costs ~1.4x on conv-heavy training, ROCm 7.2
    # XNACK servicing — profiles/RESULTS.md §18).
    # "gate-only": real VRAM + scheduler arbitration + allocation cap
    # only (zero measured overhead; no oversubscription) — for fleets
    # whose co-located working sets always fit in 288 GB.
    client_mode: str = os.environ.get("NVSHARE_CLIENT_MODE", "managed")


class DevicePluginServicer:
    """Implements v1beta1.DevicePlugin."""

    def __init__(self, cfg: PluginConfig):
        self.cfg = cfg
        self._stop = threading.Event()
        self._update = threading.Event()

    # -- rpc handlers -------------------------------------------------
    def GetDevicePluginOptions(self, request, context):
        return pb.DevicePluginOptions(pre_start_required=False)

    def _device_list(self):
        devs = []
        for gpu in self.cfg.gpus:
            for i in range(self.cfg.virtual_devices):
                devs.append(pb.Device(ID=device_id(gpu, i),
                                      health=HEALTHY))
        return devs

    def ListAndWatch(self, request, context):
        # All-healthy stub, like the reference (server.go:204-213):
        # send once, then hold the stream open re-sending on updates.
        yield pb.ListAndWatchResponse(devices=self._device_list())
        while not self._stop.is_set():
            if self._update.wait(timeout=1.0):
                self._update.clear()
                yield pb.ListAndWatchResponse(
                    devices=self._device_list())

    def Allocate(self, request, context):
        resp = pb.AllocateResponse()
        for creq in request.container_requests:
            cresp = resp.container_responses.add()
            gpus = set()
            for dev_id in creq.devicesIDs:
                gpu, _ = parse_device_id(dev_id)
                gpus.add(gpu)
            if not gpus:
                gpus = {0}
            gpu = sorted(gpus)[0]

            cresp.envs["LD_PRELOAD"] = os.path.join(
                self.cfg.lib_dir, "libnvshare.so")
            cresp.envs["HSA_XNACK"] = "1"
            cresp.envs["NVSHARE_SOCK_DIR"] = self.cfg.sock_dir + "/"
            # The node-level GPU index: inside the container the GPU
            # is always HIP device 0, so the client needs this to
            # register on the right per-GPU arbitration queue
            # (src/client.c detect_physical_gpu).
            cresp.envs["NVSHARE_GPU"] = str(gpu)
            if self.cfg.client_mode == "gate-only":
                cresp.envs["NVSHARE_DISABLE_UM"] = "1"
            if self.cfg.debug_clients:
                cresp.envs["NVSHARE_DEBUG"] = "1"

            # AMD GPU exposure: /dev/kfd (compute) + the GPU's DRI nodes.
            kfd = cresp.devices.add()
            kfd.container_path = "/dev/kfd"
            kfd.host_path = "/dev/kfd"
            kfd.permissions = "rw"
            nodes = render_nodes()
            if gpu < len(nodes):
                rd = cresp.devices.add()
                rd.container_path = nodes[gpu]
                rd.host_path = nodes[gpu]
                rd.permissions = "rw"

            m = cresp.mounts.add()
            m.container_path = os.path.join(self.cfg.lib_dir,
                                            "libnvshare.so")
            m.host_path = os.path.join(self.cfg.sock_dir,
                                       "libnvshare.so")
            m.read_only = True
            m = cresp.mounts.add()
            m.container_path = self.cfg.sock_dir
            m.host_path = self.cfg.sock_dir
            m.read_only = False
            log.info("Allocate: ids=%s -> gpu%d",
                     list(creq.devicesIDs), gpu)
        return resp

    def GetPreferredAllocation(self, request, context):
        resp = pb.PreferredAllocationResponse()
        for creq in request.container_requests:
            cresp = resp.container_responses.add()
            cresp.deviceIDs.extend(
                list(creq.available_deviceIDs)[:creq.allocation_size])
        return resp

    def PreStartContainer(self, request, context):
        return pb.PreStartContainerResponse()

    def stop(self):
        self._stop.set()
        self._update.set()


def _handler(servicer):
    rpcs = {
        "GetDevicePluginOptions": grpc.unary_unary_rpc_method_handler(
            lambda req, ctx: servicer.GetDevicePluginOptions(
                pb.Empty.FromString(req), ctx),
            response_serializer=lambda m: m.SerializeToString()),
        "ListAndWatch": grpc.unary_stream_rpc_method_handler(
            lambda req, ctx: servicer.ListAndWatch(
                pb.Empty.FromString(req), ctx),
            response_serializer=lambda m: m.SerializeToString()),
        "Allocate": grpc.unary_unary_rpc_method_handler(
            lambda req, ctx: servicer.Allocate(
                pb.AllocateRequest.FromString(req), ctx),
            response_serializer=lambda m: m.SerializeToString()),
        "GetPreferredAllocation": grpc.unary_unary_rpc_method_handler(
            lambda req, ctx: servicer.GetPreferredAllocation(
                pb.PreferredAllocationRequest.FromString(req), ctx),
            response_serializer=lambda m: m.SerializeToString()),
        "PreStartContainer": grpc.unary_unary_rpc_method_handler(
            lambda req, ctx: servicer.PreStartContainer(
                pb.PreStartContainerRequest.FromString(req), ctx),
            response_serializer=lambda m: m.SerializeToString()),
    }
    return grpc.method_handlers_generic_handler(
        pb.DEVICE_PLUGIN_SERVICE, rpcs)


class DevicePlugin:
    """Server lifecycle: serve on our socket, register with kubelet,
    restart when the kubelet socket is recreated (reference
    main.go:132-177)."""

    def __init__(self, cfg: PluginConfig | None = None):
        self.cfg = cfg or PluginConfig()
        self.servicer = DevicePluginServicer(self.cfg)
        self.server: grpc.Server | None = None
        self._shutdown = threading.Event()

    @property
    def endpoint(self) -> str:
        return os.path.join(self.cfg.kubelet_dir, PLUGIN_SOCK)

    def start(self) -> None:
        sock = self.endpoint
        if os.path.exists(sock):
            os.unlink(sock)
        self.server = grpc.server(
            futures.ThreadPoolExecutor(max_workers=8))
        self.server.add_generic_rpc_handlers((_handler(self.servicer),))
        self.server.add_insecure_port(f"unix://{sock}")
        self.server.start()
        log.info("device plugin serving on %s (%d GPUs x %d virtual)",
                 sock, len(self.cfg.gpus), self.cfg.virtual_devices)

    def register(self, timeout: float = 10.0) -> None:
        kubelet = os.path.join(self.cfg.kubelet_dir, KUBELET_SOCK)
        chan = grpc.insecure_channel(f"unix://{kubelet}")
        register = chan.unary_unary(
            f"/{pb.REGISTRATION_SERVICE}/Register",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=pb.Empty.FromString)
        req = pb.RegisterRequest(
            version=pb.API_VERSION,
            endpoint=PLUGIN_SOCK,
            resource_name=self.cfg.resource_name,
            options=pb.DevicePluginOptions(pre_start_required=False))
        register(req, timeout=timeout)
        chan.close()
        log.info("registered %s with kubelet", self.cfg.resource_name)

    def stop(self) -> None:
        self._shutdown.set()
        self.servicer.stop()
        if self.server is not None:
            self.server.stop(grace=1)
            self.server = None

    def run_forever(self) -> None:
        """Serve + register; restart on kubelet socket recreation."""
        kubelet = os.path.join(self.cfg.kubelet_dir, KUBELET_SOCK)
        while not self._shutdown.is_set():
            self.start()
            try:
                self.register(timeout=5)
            except Exception as e:  # noqa: BLE001 — keep serving loop alive
                log.error("kubelet registration failed: %s", e)
                self.servicer.stop()
                if self.server is not None:
                    self.server.stop(grace=1)
                    self.server = None
                self._shutdown.wait(2)
                continue
            def socket_id():
                # inode numbers get reused quickly on tmpfs; ctime
                # disambiguates a same-inode recreation.
                try:
                    st = os.stat(kubelet)
                    return (st.st_ino, st.st_ctime_ns)
                except FileNotFoundError:
                    return None

            ino = socket_id()
            while not self._shutdown.is_set():
                if self._shutdown.wait(2):
                    break
                if socket_id() != ino:
                    log.warning("kubelet socket changed; restarting")
                    break
            if not self._shutdown.is_set():
                self.servicer.stop()
                if self.server is not None:
                    self.server.stop(grace=1)
                    self.server = None
                # a fresh servicer for the next serve cycle
                self.servicer = DevicePluginServicer(self.cfg)


def main() -> None:
    logging.basicConfig(
        level=logging.INFO,
        format="[NVSHARE][device-plugin] %(message)s")
    DevicePlugin().run_forever()


if __name__ == "__main__":
    main()
