"""ctypes wrapper for the gfx950 utility kernels (hip/hiputil.hip).

GPU-only: importing is safe anywhere, calling load() requires a box
with an MI355X and the built libnvshare_hiputil.so.
"""

from __future__ import annotations

import ctypes
from dataclasses import dataclass

from nvshare_amd.paths import artifacts


@dataclass
class HipUtil:
    lib: ctypes.CDLL

    def _ck(self, rc: int, what: str) -> None:
        if rc != 0:
            raise RuntimeError(f"{what} failed with hipError_t={rc}")

    def malloc_managed(self, nbytes: int) -> int:
        p = ctypes.c_void_p()
        self._ck(self.lib.nvs_hip_malloc_managed(ctypes.byref(p),
                                                 ctypes.c_size_t(nbytes)),
                 "hipMallocManaged")
        assert p.value is not None
        return p.value

    def malloc(self, nbytes: int) -> int:
        p = ctypes.c_void_p()
        self._ck(self.lib.nvs_hip_malloc(ctypes.byref(p),
                                         ctypes.c_size_t(nbytes)),
                 "hipMalloc")
        assert p.value is not None
        return p.value

    def free(self, ptr: int) -> None:
        self._ck(self.lib.nvs_hip_free(ctypes.c_void_p(ptr)), "hipFree")

    def prefetch(self, ptr: int, nbytes: int, device: int = 0) -> None:
        self._ck(self.lib.nvs_hip_prefetch(ctypes.c_void_p(ptr),
                                           ctypes.c_size_t(nbytes),
                                           device), "hipMemPrefetchAsync")

    ADVISE_PREFERRED_LOCATION = 3
    ADVISE_ACCESSED_BY = 5
    ADVISE_COARSE_GRAIN = 100

    def prefetch_chunked(self, ptr: int, nbytes: int, device: int = 0,
                         n_streams: int = 4,
                         chunk_bytes: int = 256 << 20) -> None:
        self._ck(self.lib.nvs_hip_prefetch_chunked(
            ctypes.c_void_p(ptr), ctypes.c_size_t(nbytes), device,
            n_streams, ctypes.c_size_t(chunk_bytes)),
            "prefetch_chunked")

    def advise(self, ptr: int, nbytes: int, advice: int,
               device: int = 0) -> None:
        self._ck(self.lib.nvs_hip_advise(ctypes.c_void_p(ptr),
                                         ctypes.c_size_t(nbytes),
                                         advice, device), "hipMemAdvise")

    def touch_pages(self, ptr: int, n_floats: int, stride: int = 1,
                    val: float = 1.0) -> None:
        self._ck(self.lib.nvs_touch_pages(ctypes.c_void_p(ptr),
                                          ctypes.c_size_t(n_floats),
                                          ctypes.c_size_t(stride),
                                          ctypes.c_float(val), None),
                 "touch_pages")

    def busy(self, ms: float) -> None:
        self._ck(self.lib.nvs_busy(ctypes.c_double(ms), None), "busy")

    def stream_triad(self, a: int, b: int, c: int, s: float,
                     n_floats: int) -> None:
        self._ck(self.lib.nvs_stream_triad(
            ctypes.c_void_p(a), ctypes.c_void_p(b), ctypes.c_void_p(c),
            ctypes.c_float(s), ctypes.c_size_t(n_floats), None),
            "stream_triad")

    def sync(self) -> None:
        self._ck(self.lib.nvs_hip_sync(), "hipDeviceSynchronize")

    def mem_get_info(self) -> tuple[int, int]:
        f = ctypes.c_size_t()
        t = ctypes.c_size_t()
        self._ck(self.lib.nvs_hip_mem_get_info(ctypes.byref(f),
                                               ctypes.byref(t)),
                 "hipMemGetInfo")
        return f.value, t.value


def load() -> HipUtil:
    path = artifacts().hiputil
    if not path.exists():
        raise FileNotFoundError(
            f"{path} missing; build with `make -C hip` (hipcc, gfx950)")
    lib = ctypes.CDLL(str(path))
    for fname in ("nvs_hip_malloc_managed", "nvs_hip_malloc",
                  "nvs_hip_free", "nvs_hip_prefetch", "nvs_hip_advise",
                  "nvs_hip_prefetch_chunked",
                  "nvs_touch_pages", "nvs_read_pages", "nvs_busy",
                  "nvs_stream_triad", "nvs_hip_sync",
                  "nvs_hip_mem_get_info"):
        getattr(lib, fname).restype = ctypes.c_int
    return HipUtil(lib)
