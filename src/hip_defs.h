/*
 * nvshare-amd — minimal hand-declared HIP runtime API surface.
 *
 * The interposer deliberately does not include ROCm headers: it only
 * needs the ABI of the ~30 entry points it hooks, declared here by hand
 * (the same tactic as the reference's cuda_defs.h, re-derived for HIP
 * from /opt/rocm/include/hip/hip_runtime_api.h + hip_ext.h signatures).
 * This keeps libnvshare.so dependency-free: it never links libamdhip64,
 * it resolves the real entry points at runtime with dlsym/dlopen.
 */
#ifndef NVSHARE_HIP_DEFS_H
#define NVSHARE_HIP_DEFS_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef int nvshipError_t;          /* hipError_t */
#define NVSHIP_SUCCESS 0            /* hipSuccess */
#define NVSHIP_ERROR_OOM 2          /* hipErrorOutOfMemory */
#define NVSHIP_ERROR_NOT_SUPPORTED 801
#define NVSHIP_ERROR_INVALID_VALUE 1

typedef void *nvship_stream_t;      /* hipStream_t */
typedef void *nvship_event_t;       /* hipEvent_t */
typedef void *nvship_function_t;    /* hipFunction_t */
typedef void *nvship_mempool_t;     /* hipMemPool_t */
typedef void *nvship_graphexec_t;   /* hipGraphExec_t */
typedef void *nvship_deviceptr_t;   /* hipDeviceptr_t */

typedef struct {
	unsigned int x, y, z;
} nvship_dim3;                      /* dim3 */

typedef int nvship_memcpy_kind;     /* hipMemcpyKind */

#define NVSHIP_MEM_ATTACH_GLOBAL 0x01  /* hipMemAttachGlobal */
#define NVSHIP_CPU_DEVICE_ID (-1)      /* hipCpuDeviceId */
/* hipMemoryAdvise */
#define NVSHIP_MEM_ADVISE_SET_PREFERRED_LOCATION 3
#define NVSHIP_MEM_ADVISE_UNSET_PREFERRED_LOCATION 4
#define NVSHIP_MEM_ADVISE_SET_ACCESSED_BY 5

/* Function-pointer types for the real entry points we resolve. */
typedef nvshipError_t (*fn_hipMalloc)(void **, size_t);
typedef nvshipError_t (*fn_hipExtMallocWithFlags)(void **, size_t,
						  unsigned int);
typedef nvshipError_t (*fn_hipMallocManaged)(void **, size_t, unsigned int);
typedef nvshipError_t (*fn_hipMallocAsync)(void **, size_t,
					   nvship_stream_t);
typedef nvshipError_t (*fn_hipMallocFromPoolAsync)(void **, size_t,
						   nvship_mempool_t,
						   nvship_stream_t);
typedef nvshipError_t (*fn_hipFree)(void *);
typedef nvshipError_t (*fn_hipFreeAsync)(void *, nvship_stream_t);
typedef nvshipError_t (*fn_hipMemGetInfo)(size_t *, size_t *);
typedef nvshipError_t (*fn_hipMemPrefetchAsync)(const void *, size_t, int,
						nvship_stream_t);
typedef nvshipError_t (*fn_hipMemAdvise)(const void *, size_t, int, int);
typedef nvshipError_t (*fn_hipDeviceSynchronize)(void);
typedef nvshipError_t (*fn_hipSetDevice)(int);
typedef nvshipError_t (*fn_hipGetDevice)(int *);
typedef nvshipError_t (*fn_hipStreamSynchronize)(nvship_stream_t);
typedef nvshipError_t (*fn_hipStreamCreateWithFlags)(nvship_stream_t *,
						     unsigned int);
typedef nvshipError_t (*fn_hipStreamDestroy)(nvship_stream_t);
#define NVSHIP_STREAM_NON_BLOCKING 0x01
typedef nvshipError_t (*fn_hipLaunchKernel)(const void *, nvship_dim3,
					    nvship_dim3, void **, size_t,
					    nvship_stream_t);
typedef nvshipError_t (*fn_hipExtLaunchKernel)(const void *, nvship_dim3,
					       nvship_dim3, void **, size_t,
					       nvship_stream_t,
					       nvship_event_t,
					       nvship_event_t, int);
typedef nvshipError_t (*fn_hipLaunchCooperativeKernel)(const void *,
						       nvship_dim3,
						       nvship_dim3, void **,
						       unsigned int,
						       nvship_stream_t);
typedef nvshipError_t (*fn_hipModuleLaunchKernel)(nvship_function_t,
	unsigned int, unsigned int, unsigned int, unsigned int,
	unsigned int, unsigned int, unsigned int, nvship_stream_t, void **,
	void **);
typedef nvshipError_t (*fn_hipExtModuleLaunchKernel)(nvship_function_t,
	uint32_t, uint32_t, uint32_t, uint32_t, uint32_t, uint32_t, size_t,
	nvship_stream_t, void **, void **, nvship_event_t, nvship_event_t,
	uint32_t);
typedef nvshipError_t (*fn_hipGraphLaunch)(nvship_graphexec_t,
					   nvship_stream_t);
typedef nvshipError_t (*fn_hipMemcpy)(void *, const void *, size_t,
				      nvship_memcpy_kind);
typedef nvshipError_t (*fn_hipMemcpyAsync)(void *, const void *, size_t,
					   nvship_memcpy_kind,
					   nvship_stream_t);
typedef nvshipError_t (*fn_hipMemcpyWithStream)(void *, const void *,
						size_t, nvship_memcpy_kind,
						nvship_stream_t);
typedef nvshipError_t (*fn_hipMemcpyHtoD)(nvship_deviceptr_t, const void *,
					  size_t);
typedef nvshipError_t (*fn_hipMemcpyDtoH)(void *, nvship_deviceptr_t,
					  size_t);
typedef nvshipError_t (*fn_hipMemcpyDtoD)(nvship_deviceptr_t,
					  nvship_deviceptr_t, size_t);
typedef nvshipError_t (*fn_hipMemcpyHtoDAsync)(nvship_deviceptr_t,
					       const void *, size_t,
					       nvship_stream_t);
typedef nvshipError_t (*fn_hipMemcpyDtoHAsync)(void *, nvship_deviceptr_t,
					       size_t, nvship_stream_t);
typedef nvshipError_t (*fn_hipMemcpyDtoDAsync)(nvship_deviceptr_t,
					       nvship_deviceptr_t, size_t,
					       nvship_stream_t);
typedef nvshipError_t (*fn_hipMemset)(void *, int, size_t);
typedef nvshipError_t (*fn_hipMemsetAsync)(void *, int, size_t,
					   nvship_stream_t);
typedef nvshipError_t (*fn_hipMemsetD32Async)(nvship_deviceptr_t, int,
					      size_t, nvship_stream_t);
typedef nvshipError_t (*fn_hipMemcpy2D)(void *, size_t, const void *,
					size_t, size_t, size_t,
					nvship_memcpy_kind);
typedef nvshipError_t (*fn_hipMemcpy2DAsync)(void *, size_t, const void *,
					     size_t, size_t, size_t,
					     nvship_memcpy_kind,
					     nvship_stream_t);
typedef nvshipError_t (*fn_hipMemcpyToSymbol)(const void *, const void *,
					      size_t, size_t,
					      nvship_memcpy_kind);
typedef nvshipError_t (*fn_hipMemcpyFromSymbol)(void *, const void *,
						size_t, size_t,
						nvship_memcpy_kind);
typedef nvshipError_t (*fn_hipMemcpyPeerAsync)(void *, int, const void *,
					       int, size_t,
					       nvship_stream_t);
typedef nvshipError_t (*fn_hipModuleLaunchCooperativeKernel)(
	nvship_function_t, unsigned int, unsigned int, unsigned int,
	unsigned int, unsigned int, unsigned int, unsigned int,
	nvship_stream_t, void **);
typedef nvshipError_t (*fn_hipGetProcAddress)(const char *, void **, int,
					      uint64_t, void *);
typedef const char *(*fn_hipGetErrorString)(nvshipError_t);

/* Virtual-memory-management API (PyTorch expandable segments).
 * hipMemGenericAllocationHandle_t is an opaque unsigned long long;
 * hipMemAllocationProp is opaque to us (const pointer pass-through). */
typedef unsigned long long nvship_memhandle_t;
typedef nvshipError_t (*fn_hipMemCreate)(nvship_memhandle_t *, size_t,
					 const void *, unsigned long long);
typedef nvshipError_t (*fn_hipMemRelease)(nvship_memhandle_t);
typedef nvshipError_t (*fn_hipMemMap)(void *, size_t, size_t,
				      nvship_memhandle_t,
				      unsigned long long);
typedef nvshipError_t (*fn_hipMemUnmap)(void *, size_t);

/* ROCm SMI (librocm_smi64.so), for idle detection. */
typedef int (*fn_rsmi_init)(uint64_t);
typedef int (*fn_rsmi_dev_busy_percent_get)(uint32_t, uint32_t *);

#ifdef __cplusplus
}
#endif

#endif /* NVSHARE_HIP_DEFS_H */
