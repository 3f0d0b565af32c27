"""Locate nvshare-amd build artifacts.

The native components live in-tree under src/build (built by
`make -C src`); installations can override with NVSHARE_PREFIX.
"""

from __future__ import annotations

import os
import subprocess
from dataclasses import dataclass
from pathlib import Path

REPO_ROOT = Path(__file__).resolve().parent.parent
SRC_DIR = REPO_ROOT / "src"
BUILD_DIR = SRC_DIR / "build"
HIP_BUILD_DIR = REPO_ROOT / "hip" / "build"


@dataclass(frozen=True)
class Artifacts:
    libnvshare: Path
    scheduler: Path
    ctl: Path
    stub_dir: Path        # directory holding the CPU stub libamdhip64
    hipclient: Path
    hiputil: Path         # gfx950 utility kernel library (may not exist)

    def built(self) -> bool:
        return (
            self.libnvshare.exists()
            and self.scheduler.exists()
            and self.ctl.exists()
        )


def artifacts() -> Artifacts:
    prefix = os.environ.get("NVSHARE_PREFIX")
    base = Path(prefix) if prefix else BUILD_DIR
    return Artifacts(
        libnvshare=base / "libnvshare.so",
        scheduler=base / "nvshare-scheduler",
        ctl=base / "nvsharectl",
        stub_dir=base,
        hipclient=base / "hipclient",
        hiputil=HIP_BUILD_DIR / "libnvshare_hiputil.so",
    )


def build_native(jobs: int = 0) -> None:
    """Build the native components in-tree (host-only C, seconds)."""
    cmd = ["make", "-C", str(SRC_DIR), "-s"]
    if jobs:
        cmd.append(f"-j{jobs}")
    subprocess.run(cmd, check=True)


def ensure_built() -> Artifacts:
    art = artifacts()
    if not art.built():
        build_native()
        art = artifacts()
    if not art.built():
        raise RuntimeError(
            "nvshare-amd native components missing; run `make -C src`"
        )
    return art
