"""nvshare-amd: transparent MI355X GPU sharing.

A brand-new MI355X-native framework with the capabilities of
grgalex/nvshare (reference: /root/reference): N unmodified HIP/ROCm
processes share one physical GPU, each seeing the whole 288 GB of
HBM3E, with device allocations transparently rewritten to
hipMallocManaged (gfx950 XNACK/HMM demand paging) and GPU work
serialized by an FCFS time-quantum scheduler to prevent page-fault
thrashing under oversubscription.

Components:
  - libnvshare.so        LD_PRELOAD interposer (src/hook.c, src/client.c)
  - nvshare-scheduler    per-node arbiter daemon (src/scheduler.c)
  - nvsharectl           config CLI (src/ctl.c)
  - nvshare_amd (this)   python control plane, workloads, K8s device plugin
"""

__version__ = "0.1.0"

from nvshare_amd.paths import artifacts  # noqa: F401
from nvshare_amd.scheduler import SchedulerDaemon  # noqa: F401
from nvshare_amd.env import client_env  # noqa: F401
from nvshare_amd import ctl  # noqa: F401
