/*
 * nvshare-amd — in-app client runtime (scheduling agent).
 *
 * Linked into libnvshare.so. Two daemon threads are injected into the
 * application: the client thread (persistent scheduler connection,
 * LOCK_OK/DROP_LOCK handling) and the early-release thread (idle
 * detection). Behavioral parity reference: /root/reference/src/client.{c,h}.
 */
#ifndef NVSHARE_CLIENT_H
#define NVSHARE_CLIENT_H

#include <stddef.h>

#include "hip_defs.h"

/* Real HIP entry points, resolved by hook.c's bootstrap. */
struct nvs_real_hip {
	fn_hipMalloc hipMalloc;
	fn_hipExtMallocWithFlags hipExtMallocWithFlags;
	fn_hipMallocManaged hipMallocManaged;
	fn_hipMallocAsync hipMallocAsync;
	fn_hipMallocFromPoolAsync hipMallocFromPoolAsync;
	fn_hipFree hipFree;
	fn_hipFreeAsync hipFreeAsync;
	fn_hipMemGetInfo hipMemGetInfo;
	fn_hipMemPrefetchAsync hipMemPrefetchAsync;
	fn_hipMemAdvise hipMemAdvise;
	fn_hipDeviceSynchronize hipDeviceSynchronize;
	fn_hipSetDevice hipSetDevice;
	fn_hipGetDevice hipGetDevice;
	fn_hipStreamSynchronize hipStreamSynchronize;
	fn_hipStreamCreateWithFlags hipStreamCreateWithFlags;
	fn_hipStreamDestroy hipStreamDestroy;
	fn_hipLaunchKernel hipLaunchKernel;
	fn_hipExtLaunchKernel hipExtLaunchKernel;
	fn_hipLaunchCooperativeKernel hipLaunchCooperativeKernel;
	fn_hipModuleLaunchKernel hipModuleLaunchKernel;
	fn_hipExtModuleLaunchKernel hipExtModuleLaunchKernel;
	fn_hipGraphLaunch hipGraphLaunch;
	fn_hipMemcpy hipMemcpy;
	fn_hipMemcpyAsync hipMemcpyAsync;
	fn_hipMemcpyWithStream hipMemcpyWithStream;
	fn_hipMemcpyHtoD hipMemcpyHtoD;
	fn_hipMemcpyDtoH hipMemcpyDtoH;
	fn_hipMemcpyDtoD hipMemcpyDtoD;
	fn_hipMemcpyHtoDAsync hipMemcpyHtoDAsync;
	fn_hipMemcpyDtoHAsync hipMemcpyDtoHAsync;
	fn_hipMemcpyDtoDAsync hipMemcpyDtoDAsync;
	fn_hipMemset hipMemset;
	fn_hipMemsetAsync hipMemsetAsync;
	fn_hipMemsetD32Async hipMemsetD32Async;
	fn_hipMemcpy2D hipMemcpy2D;
	fn_hipMemcpy2DAsync hipMemcpy2DAsync;
	fn_hipMemcpyToSymbol hipMemcpyToSymbol;
	fn_hipMemcpyFromSymbol hipMemcpyFromSymbol;
	fn_hipMemcpyPeerAsync hipMemcpyPeerAsync;
	fn_hipModuleLaunchCooperativeKernel hipModuleLaunchCooperativeKernel;
	fn_hipGetProcAddress hipGetProcAddress;
	fn_hipGetErrorString hipGetErrorString;
	fn_hipMemCreate hipMemCreate;
	fn_hipMemRelease hipMemRelease;
	fn_hipMemMap hipMemMap;
	fn_hipMemUnmap hipMemUnmap;
};

extern struct nvs_real_hip real;

/* Spawns the client + early-release threads; blocks until the scheduler
 * handshake finishes (or standalone fallback engages). Called exactly
 * once from the hook bootstrap. */
void nvs_client_init(void);

/* The hot gate: block the calling app thread until this process may
 * submit GPU work. Returns with the submission read-lock HELD; the
 * caller must call nvs_submit_end() after the real HIP call returns. */
void nvs_submit_begin(void);
void nvs_submit_end(void);

/* The device the app last selected (for drains from client threads). */
extern int nvs_app_device;

/* 1 while a scheduler is actively gating this process. */
int nvs_scheduler_gating(void);

/* 1 when this process may submit GPU work right now (racy read). */
int nvs_can_submit_now(void);

/* Populate managed allocations whose eager prefetch was deferred
 * because we didn't hold the GPU lock at hipMalloc time (hook.c). */
void nvs_populate_pending(void);

/* Managed-allocation registry hooks (hook.c) used for prefetch. Called
 * with the list snapshot under the allocation lock. */
void nvs_prefetch_allocs(void);

/* Migrate the tracked working set to host after a lock release
 * (NVSHARE_EVICT=1). */
void nvs_evict_allocs(void);

/* Sum of tracked allocations in MiB (hook.c). */
long nvs_sum_allocated_mib(void);

/* Advertised device capacity in MiB (hook.c). */
long nvs_mem_total_mib(void);

/* Drop the managed free cache without releasing (fork child). */
void nvs_free_cache_forget(void);

#endif /* NVSHARE_CLIENT_H */
