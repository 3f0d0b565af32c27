/*
 * CPU stub of libamdhip64 for hook-level tests without a GPU.
 *
 * Implements the interposed HIP surface over plain malloc and logs
 * every call to the file named by NVSTUB_LOG (one line per event:
 * "<ns> <pid> <event> <arg>"), so tests can assert (a) that the
 * interposer rewrote hipMalloc into hipMallocManaged, and (b) that
 * co-located clients' kernel windows never overlap.  Kernel launches
 * sleep NVSTUB_KERNEL_US microseconds to emulate GPU work.
 * Advertised memory: NVSTUB_TOTAL_MIB (default 1024).
 *
 * This mirrors the test strategy SURVEY.md §4 calls for (the reference
 * had no such harness).  Test-only: never shipped.
 */
#define _GNU_SOURCE
#include <fcntl.h>
#include <pthread.h>
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/file.h>
#include <time.h>
#include <unistd.h>

typedef int hipError_t;
typedef void *hipStream_t;
typedef struct { unsigned x, y, z; } dim3_t;

static pthread_mutex_t log_mutex = PTHREAD_MUTEX_INITIALIZER;
static int log_fd = -2; /* -2 = uninitialized, -1 = disabled */
static long kernel_us;
static long sync_us;
static size_t total_bytes;
static size_t used_bytes;
static pthread_mutex_t mem_mutex = PTHREAD_MUTEX_INITIALIZER;

static int64_t now_ns(void)
{
	struct timespec ts;
	clock_gettime(CLOCK_REALTIME, &ts);
	return (int64_t)ts.tv_sec * 1000000000LL + ts.tv_nsec;
}

static void stub_init(void)
{
	const char *v;

	if (log_fd != -2)
		return;
	pthread_mutex_lock(&log_mutex);
	if (log_fd == -2) {
		v = getenv("NVSTUB_LOG");
		log_fd = (v != NULL) ?
			open(v, O_WRONLY | O_CREAT | O_APPEND, 0666) : -1;
		v = getenv("NVSTUB_KERNEL_US");
		kernel_us = v != NULL ? atol(v) : 0;
		v = getenv("NVSTUB_SYNC_US");
		sync_us = v != NULL ? atol(v) : 0;
		v = getenv("NVSTUB_TOTAL_MIB");
		total_bytes = (v != NULL ? (size_t)atol(v) : 1024)
			* 1024 * 1024;
	}
	pthread_mutex_unlock(&log_mutex);
}

static void ev(const char *name, long long arg)
{
	char buf[128];
	int n;

	stub_init();
	if (log_fd < 0)
		return;
	n = snprintf(buf, sizeof(buf), "%lld %d %s %lld\n",
		     (long long)now_ns(), (int)getpid(), name, arg);
	pthread_mutex_lock(&log_mutex);
	if (write(log_fd, buf, (size_t)n) != n) { /* best effort */ }
	pthread_mutex_unlock(&log_mutex);
}

static hipError_t do_alloc(void **ptr, size_t size)
{
	pthread_mutex_lock(&mem_mutex);
	used_bytes += size;
	pthread_mutex_unlock(&mem_mutex);
	*ptr = malloc(size < 16 ? 16 : size);
	return *ptr != NULL ? 0 : 2 /* hipErrorOutOfMemory */;
}

hipError_t hipMalloc(void **ptr, size_t size)
{
	ev("hipMalloc", (long long)size);
	return do_alloc(ptr, size);
}

hipError_t hipMallocManaged(void **ptr, size_t size, unsigned int flags)
{
	(void)flags;
	ev("hipMallocManaged", (long long)size);
	return do_alloc(ptr, size);
}

hipError_t hipExtMallocWithFlags(void **ptr, size_t size, unsigned int f)
{
	(void)f;
	ev("hipExtMallocWithFlags", (long long)size);
	return do_alloc(ptr, size);
}

hipError_t hipMallocAsync(void **ptr, size_t size, hipStream_t s)
{
	(void)s;
	ev("hipMallocAsync", (long long)size);
	return do_alloc(ptr, size);
}

hipError_t hipMallocFromPoolAsync(void **ptr, size_t size, void *pool,
				  hipStream_t s)
{
	(void)pool; (void)s;
	ev("hipMallocFromPoolAsync", (long long)size);
	return do_alloc(ptr, size);
}

/* VMM API (expandable segments): handle = the malloc'd block. */
hipError_t hipMemCreate(unsigned long long *handle, size_t size,
			const void *prop, unsigned long long flags)
{
	void *p = NULL;
	hipError_t r;

	(void)prop; (void)flags;
	ev("hipMemCreate", (long long)size);
	r = do_alloc(&p, size);
	*handle = (unsigned long long)(uintptr_t)p;
	return r;
}

hipError_t hipMemRelease(unsigned long long handle)
{
	ev("hipMemRelease", (long long)handle);
	free((void *)(uintptr_t)handle);
	return 0;
}

hipError_t hipMemMap(void *ptr, size_t size, size_t offset,
		     unsigned long long handle, unsigned long long flags)
{
	(void)ptr; (void)size; (void)offset; (void)handle; (void)flags;
	ev("hipMemMap", (long long)size);
	return 0;
}

hipError_t hipFree(void *ptr)
{
	ev("hipFree", (long long)(uintptr_t)ptr);
	free(ptr);
	return 0;
}

hipError_t hipFreeAsync(void *ptr, hipStream_t s)
{
	(void)s;
	ev("hipFreeAsync", (long long)(uintptr_t)ptr);
	free(ptr);
	return 0;
}

hipError_t hipMemGetInfo(size_t *freep, size_t *totalp)
{
	stub_init();
	ev("hipMemGetInfo", 0);
	if (totalp)
		*totalp = total_bytes;
	if (freep)
		*freep = used_bytes < total_bytes ?
			total_bytes - used_bytes : 0;
	return 0;
}

hipError_t hipMemPrefetchAsync(const void *p, size_t n, int dev,
			       hipStream_t s)
{
	(void)p; (void)s;
	/* Direction-tagged so tests can tell eviction (to CPU, dev -1)
	 * from restore (to device). */
	ev(dev < 0 ? "hipMemPrefetchAsync_cpu" : "hipMemPrefetchAsync",
	   (long long)n);
	return 0;
}

hipError_t hipMemAdvise(const void *p, size_t n, int advice, int dev)
{
	(void)p; (void)advice; (void)dev;
	ev("hipMemAdvise", (long long)n);
	return 0;
}

hipError_t hipDeviceSynchronize(void)
{
	stub_init();
	ev("hipDeviceSynchronize", 0);
	if (sync_us > 0)
		usleep((useconds_t)sync_us);
	return 0;
}

hipError_t hipSetDevice(int dev)
{
	ev("hipSetDevice", dev);
	return 0;
}

hipError_t hipGetDevice(int *dev)
{
	if (dev)
		*dev = 0;
	return 0;
}

hipError_t hipStreamSynchronize(hipStream_t s)
{
	(void)s;
	ev("hipStreamSynchronize", 0);
	return 0;
}

hipError_t hipStreamCreateWithFlags(hipStream_t *s, unsigned int flags)
{
	(void)flags;
	if (s)
		*s = (hipStream_t)0x1;
	return 0;
}

hipError_t hipStreamDestroy(hipStream_t s)
{
	(void)s;
	return 0;
}

static hipError_t run_kernel(void)
{
	stub_init();
	ev("launch_begin", kernel_us);
	if (kernel_us > 0)
		usleep((useconds_t)kernel_us);
	ev("launch_end", kernel_us);
	return 0;
}

hipError_t hipLaunchKernel(const void *f, dim3_t g, dim3_t b, void **args,
			   size_t shmem, hipStream_t s)
{
	(void)f; (void)g; (void)b; (void)args; (void)shmem; (void)s;
	return run_kernel();
}

hipError_t hipExtLaunchKernel(const void *f, dim3_t g, dim3_t b,
			      void **args, size_t shmem, hipStream_t s,
			      void *e0, void *e1, int flags)
{
	(void)f; (void)g; (void)b; (void)args; (void)shmem; (void)s;
	(void)e0; (void)e1; (void)flags;
	return run_kernel();
}

hipError_t hipLaunchCooperativeKernel(const void *f, dim3_t g, dim3_t b,
				      void **args, unsigned int shmem,
				      hipStream_t s)
{
	(void)f; (void)g; (void)b; (void)args; (void)shmem; (void)s;
	return run_kernel();
}

hipError_t hipModuleLaunchKernel(void *f, unsigned gx, unsigned gy,
				 unsigned gz, unsigned bx, unsigned by,
				 unsigned bz, unsigned shmem,
				 hipStream_t s, void **params, void **extra)
{
	(void)f; (void)gx; (void)gy; (void)gz; (void)bx; (void)by;
	(void)bz; (void)shmem; (void)s; (void)params; (void)extra;
	return run_kernel();
}

hipError_t hipExtModuleLaunchKernel(void *f, uint32_t gwx, uint32_t gwy,
				    uint32_t gwz, uint32_t lwx,
				    uint32_t lwy, uint32_t lwz,
				    size_t shmem, hipStream_t s,
				    void **params, void **extra, void *e0,
				    void *e1, uint32_t flags)
{
	(void)f; (void)gwx; (void)gwy; (void)gwz; (void)lwx; (void)lwy;
	(void)lwz; (void)shmem; (void)s; (void)params; (void)extra;
	(void)e0; (void)e1; (void)flags;
	return run_kernel();
}

hipError_t hipGraphLaunch(void *g, hipStream_t s)
{
	(void)g; (void)s;
	return run_kernel();
}

static hipError_t do_copy(void *dst, const void *src, size_t n)
{
	ev("memcpy", (long long)n);
	if (dst != NULL && src != NULL && n > 0)
		memmove(dst, src, n);
	return 0;
}

hipError_t hipMemcpy(void *d, const void *s, size_t n, int k)
{
	(void)k;
	return do_copy(d, s, n);
}

hipError_t hipMemcpyAsync(void *d, const void *s, size_t n, int k,
			  hipStream_t st)
{
	(void)k; (void)st;
	return do_copy(d, s, n);
}

hipError_t hipMemcpyWithStream(void *d, const void *s, size_t n, int k,
			       hipStream_t st)
{
	(void)k; (void)st;
	return do_copy(d, s, n);
}

hipError_t hipMemcpyHtoD(void *d, const void *s, size_t n)
{
	return do_copy(d, s, n);
}

hipError_t hipMemcpyDtoH(void *d, void *s, size_t n)
{
	return do_copy(d, s, n);
}

hipError_t hipMemcpyDtoD(void *d, void *s, size_t n)
{
	return do_copy(d, s, n);
}

hipError_t hipMemcpyHtoDAsync(void *d, const void *s, size_t n,
			      hipStream_t st)
{
	(void)st;
	return do_copy(d, s, n);
}

hipError_t hipMemcpyDtoHAsync(void *d, void *s, size_t n, hipStream_t st)
{
	(void)st;
	return do_copy(d, s, n);
}

hipError_t hipMemcpyDtoDAsync(void *d, void *s, size_t n, hipStream_t st)
{
	(void)st;
	return do_copy(d, s, n);
}

hipError_t hipMemcpy2D(void *d, size_t dp, const void *s, size_t sp,
		       size_t w, size_t h, int k)
{
	(void)dp; (void)sp; (void)k;
	return do_copy(d, s, w * h);
}

hipError_t hipMemcpy2DAsync(void *d, size_t dp, const void *s, size_t sp,
			    size_t w, size_t h, int k, hipStream_t st)
{
	(void)dp; (void)sp; (void)k; (void)st;
	return do_copy(d, s, w * h);
}

hipError_t hipMemcpyToSymbol(const void *sym, const void *s, size_t n,
			     size_t off, int k)
{
	(void)sym; (void)s; (void)n; (void)off; (void)k;
	ev("memcpyToSymbol", (long long)n);
	return 0;
}

hipError_t hipMemcpyFromSymbol(void *d, const void *sym, size_t n,
			       size_t off, int k)
{
	(void)d; (void)sym; (void)n; (void)off; (void)k;
	ev("memcpyFromSymbol", (long long)n);
	return 0;
}

hipError_t hipMemcpyPeerAsync(void *d, int dd, const void *s, int sd,
			      size_t n, hipStream_t st)
{
	(void)dd; (void)sd; (void)st;
	return do_copy(d, s, n);
}

hipError_t hipMemsetD32Async(void *d, int v, size_t count, hipStream_t s)
{
	(void)s;
	ev("memsetD32", (long long)count);
	if (d != NULL) {
		int *p = d;
		size_t i;

		for (i = 0; i < count; i++)
			p[i] = v;
	}
	return 0;
}

hipError_t hipModuleLaunchCooperativeKernel(void *f, unsigned gx,
					    unsigned gy, unsigned gz,
					    unsigned bx, unsigned by,
					    unsigned bz, unsigned shmem,
					    hipStream_t s, void **params)
{
	(void)f; (void)gx; (void)gy; (void)gz; (void)bx; (void)by;
	(void)bz; (void)shmem; (void)s; (void)params;
	return run_kernel();
}

hipError_t hipMemset(void *d, int v, size_t n)
{
	ev("memset", (long long)n);
	if (d != NULL)
		memset(d, v, n);
	return 0;
}

hipError_t hipMemsetAsync(void *d, int v, size_t n, hipStream_t s)
{
	(void)s;
	return hipMemset(d, v, n);
}

hipError_t hipGetProcAddress(const char *symbol, void **pfn, int ver,
			     uint64_t flags, void *status)
{
	(void)ver; (void)flags; (void)status;
	extern void *dlsym(void *, const char *);
	if (pfn == NULL)
		return 1;
	*pfn = dlsym(NULL, symbol); /* RTLD_DEFAULT == NULL glibc */
	return *pfn != NULL ? 0 : 1;
}

const char *hipGetErrorString(hipError_t e)
{
	static const char *oom = "hipErrorOutOfMemory";
	static const char *ok = "hipSuccess";
	static const char *other = "hipError";

	if (e == 0)
		return ok;
	if (e == 2)
		return oom;
	return other;
}
