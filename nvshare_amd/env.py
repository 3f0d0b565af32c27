"""Build the environment for an nvshare-amd client process.

Everything a sharing client needs: LD_PRELOAD of the interposer,
HSA_XNACK=1 for gfx950 page-granular demand paging, the scheduler
socket directory, and the knobs documented in src/hook.c / src/client.c.
This is exactly what the K8s device plugin injects into pods
(reference behavior: server.go:230-272).
"""

from __future__ import annotations

import os
from typing import Mapping

from nvshare_amd.paths import ensure_built


def client_env(
    sock_dir: str | None = None,
    base: Mapping[str, str] | None = None,
    debug: bool = False,
    oversubscribe: bool = False,
    standalone: bool = False,
    reserve_mib: int | None = None,
    fake_total_mib: int | None = None,
    prefetch: bool | None = None,
    evict: bool | None = None,
    disable_um: bool = False,
    use_stub: bool = False,
    extra: Mapping[str, str] | None = None,
) -> dict[str, str]:
    art = ensure_built()
    env = dict(base if base is not None else os.environ)

    preload = str(art.libnvshare)
    if env.get("LD_PRELOAD"):
        preload = preload + ":" + env["LD_PRELOAD"]
    env["LD_PRELOAD"] = preload
    # gfx950 page-granular demand paging; an explicit operator setting
    # (e.g. HSA_XNACK=0 for legacy whole-buffer residency) wins.
    env.setdefault("HSA_XNACK", "1")

    # PyTorch's expandable-segments allocator backend goes through
    # hipMemCreate/hipMemMap: the interposer caps it (hook.c) but the
    # physical VMM handles cannot be converted to managed memory, so
    # oversubscription/migration would silently not apply.  Strip the
    # option so clients use the default (hipMalloc) backend.
    for var in ("PYTORCH_HIP_ALLOC_CONF", "PYTORCH_CUDA_ALLOC_CONF"):
        conf = env.get(var)
        if conf and "expandable_segments" in conf:
            kept = [p for p in conf.split(",")
                    if "expandable_segments" not in p]
            if kept:
                env[var] = ",".join(kept)
            else:
                env.pop(var, None)

    # Deterministic MIOpen conv-algo selection: heuristic-based FAST
    # find instead of a cold exhaustive search (a fresh box otherwise
    # runs ~50k tuning kernels inside the gated region, round-1
    # GPUTEST fresh-box hang).  Callers can override via base/extra.
    env.setdefault("MIOPEN_FIND_MODE", "FAST")

    if sock_dir:
        env["NVSHARE_SOCK_DIR"] = sock_dir
    env["NVSHARE_DEBUG"] = "1" if debug else env.get("NVSHARE_DEBUG", "0")
    if oversubscribe:
        env["NVSHARE_ENABLE_SINGLE_OVERSUB"] = "1"
    if standalone:
        env["NVSHARE_STANDALONE"] = "1"
    if reserve_mib is not None:
        env["NVSHARE_RESERVE_MIB"] = str(reserve_mib)
    if fake_total_mib is not None:
        env["NVSHARE_FAKE_TOTAL_MIB"] = str(fake_total_mib)
    if prefetch is not None:
        env["NVSHARE_PREFETCH"] = "1" if prefetch else "0"
    if evict is not None:
        env["NVSHARE_EVICT"] = "1" if evict else "0"
    if disable_um:
        env["NVSHARE_DISABLE_UM"] = "1"

    if use_stub:
        # CPU tests: make the stub libamdhip64 shadow the real one.
        env["LD_LIBRARY_PATH"] = (
            str(art.stub_dir)
            + (":" + env["LD_LIBRARY_PATH"]
               if env.get("LD_LIBRARY_PATH") else "")
        )

    if extra:
        env.update(extra)
    return env
