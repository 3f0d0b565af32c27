/*
 * libnvshare.so — pure-HIP LD_PRELOAD interposer for transparent MI355X
 * GPU sharing (nvshare-amd).
 *
 * What it does (parity reference: /root/reference/src/hook.c):
 *   1. Converts device allocations (hipMalloc & friends) into
 *      hipMallocManaged so GPU memory demand-pages over gfx950
 *      XNACK/HMM and becomes oversubscribable with host RAM as swap.
 *   2. Tracks allocations, enforces a per-process cap
 *      (total - reserve) unless NVSHARE_ENABLE_SINGLE_OVERSUB=1, and
 *      under-reports free memory by the reserve in hipMemGetInfo.
 *   3. Gates every work submission (kernel launches, memcpies, memsets,
 *      graph launches) behind the scheduler lock via the client runtime
 *      (client.c) so co-located processes never page-fault-thrash.
 *   4. Bounds the undrained submission queue with an adaptive
 *      pending-kernel window so a DROP_LOCK can be honored quickly.
 *
 * MI355X-first design notes:
 *   - ROCm merges runtime+driver in libamdhip64.so, so plain ELF symbol
 *     interposition covers PyTorch/TF-ROCm; the versioned-dlsym +
 *     hipGetProcAddress hooks below additionally catch dlopen-style
 *     loaders (the reference needed a much hairier dlsym/
 *     cuGetProcAddress double path, hook.c:418-643).
 *   - The reserve default is sized for 288 GB HBM3E (8 GiB, vs the
 *     reference's 1536 MiB for 16 GB; NVSHARE_RESERVE_MIB overrides).
 *   - NVSHARE_FAKE_TOTAL_MIB shrinks the advertised capacity so
 *     oversubscription behavior is testable without filling 288 GB.
 *   - Requires HSA_XNACK=1 for page-granular demand paging; warns
 *     loudly when unset (coarse-grained migration otherwise).
 *
 * Env vars: NVSHARE_DEBUG, NVSHARE_ENABLE_SINGLE_OVERSUB,
 *   NVSHARE_RESERVE_MIB, NVSHARE_FAKE_TOTAL_MIB, NVSHARE_DISABLE_UM,
 *   NVSHARE_WINDOW_START/MAX, NVSHARE_SYNC_SLOW_MS/VERY_SLOW_MS,
 *   plus the client-side ones documented in client.c.
 */
#define _GNU_SOURCE
#include <dlfcn.h>
#include <pthread.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/mman.h>

#include "common.h"
#include "hip_defs.h"
#include "client.h"

/* ------------------------------------------------------------------ */
/* Real-symbol resolution                                              */
/* ------------------------------------------------------------------ */

static void *libhip_handle;
static pthread_once_t bootstrap_once = PTHREAD_ONCE_INIT;

/*
 * The real (libc) dlsym.  All internal resolution MUST go through this:
 * a plain dlsym call from inside this library binds to our own exported
 * dlsym hook, which reroutes hooked names back to our wrappers —
 * infinite recursion / pthread_once self-deadlock otherwise.
 */
typedef void *(*fn_dlsym)(void *, const char *);
static fn_dlsym real_dlsym_p;
static pthread_once_t dlsym_once = PTHREAD_ONCE_INIT;

static void resolve_real_dlsym(void)
{
	real_dlsym_p = (fn_dlsym)dlvsym(RTLD_NEXT, "dlsym", "GLIBC_2.34");
	if (real_dlsym_p == NULL)
		real_dlsym_p = (fn_dlsym)dlvsym(RTLD_NEXT, "dlsym",
						"GLIBC_2.2.5");
	if (real_dlsym_p == NULL)
		real_dlsym_p = (fn_dlsym)dlvsym(RTLD_DEFAULT, "dlsym",
						"GLIBC_2.34");
}

static void *real_dlsym(void *handle, const char *name)
{
	pthread_once(&dlsym_once, resolve_real_dlsym);
	return real_dlsym_p != NULL ? real_dlsym_p(handle, name) : NULL;
}

/* allocation tracking */
struct nvs_alloc {
	void *ptr;
	size_t size;
	int populated;   /* device pages materialized (eager prefetch) */
	int bulk_ready;  /* whole range residency established by the
			  * first-bulk-write prefetch (copy_prefetch) */
	int passthrough; /* real-VRAM (stream-ordered) alloc: counted
			  * against the cap but never migrated */
	struct nvs_alloc *next;
};
static struct nvs_alloc *alloc_list;
static size_t sum_allocated;
static long n_unpopulated; /* atomic; fast-path check in the gate */
static pthread_mutex_t alloc_mutex = PTHREAD_MUTEX_INITIALIZER;

static size_t mem_total;        /* advertised total (bytes) */
static size_t mem_reserve;      /* carve-out (bytes) */
static size_t free_cache_bytes; /* held-but-freed cache (guarded by
				 * alloc_mutex; see the free-cache
				 * block below) */
static int oversub_allowed;
static int disable_um;
static int alloc_prefetch = 0;   /* NVSHARE_ALLOC_PREFETCH (see
				  * populate_managed: corrupts on ROCm
				  * 7.2, default off) */
static int coarse_grain = 1;     /* NVSHARE_COARSE_GRAIN */
static int preferred_loc = 1;    /* NVSHARE_PREFERRED_LOC */
static int alloc_memset = 0;     /* NVSHARE_ALLOC_MEMSET: populate
				  * fresh managed ranges with a
				  * device-side zero fill at alloc
				  * time (pages born in HBM; synced
				  * before the pointer is returned,
				  * so nothing can race it) */
static int thp_advise = 0;       /* NVSHARE_THP: MADV_HUGEPAGE on
				  * managed ranges (experiment knob:
				  * 2 MiB mappings cut device TLB
				  * pressure if the driver honors
				  * them) */
static pthread_once_t memquery_once = PTHREAD_ONCE_INIT;

/* pending-kernel window */
static pthread_mutex_t win_mutex = PTHREAD_MUTEX_INITIALIZER;
static long window = 16;
static long window_max = 2048;
static long kern_since_sync;
static long sync_slow_ms = 1000;
static long sync_very_slow_ms = 10000;

static void *resolve(const char *name)
{
	void *p = real_dlsym(RTLD_NEXT, name);

	if (p != NULL)
		return p;
	if (libhip_handle == NULL) {
		libhip_handle = dlopen("libamdhip64.so.7",
				       RTLD_LAZY | RTLD_GLOBAL);
		if (libhip_handle == NULL)
			libhip_handle = dlopen("libamdhip64.so",
					       RTLD_LAZY | RTLD_GLOBAL);
	}
	if (libhip_handle != NULL)
		p = real_dlsym(libhip_handle, name);
	return p;
}

static void bootstrap(void)
{
	nvs_log_init();

	real.hipMalloc = (fn_hipMalloc)resolve("hipMalloc");
	real.hipExtMallocWithFlags =
		(fn_hipExtMallocWithFlags)resolve("hipExtMallocWithFlags");
	real.hipMallocManaged =
		(fn_hipMallocManaged)resolve("hipMallocManaged");
	real.hipMallocAsync = (fn_hipMallocAsync)resolve("hipMallocAsync");
	real.hipMallocFromPoolAsync = (fn_hipMallocFromPoolAsync)
		resolve("hipMallocFromPoolAsync");
	real.hipFree = (fn_hipFree)resolve("hipFree");
	real.hipFreeAsync = (fn_hipFreeAsync)resolve("hipFreeAsync");
	real.hipMemGetInfo = (fn_hipMemGetInfo)resolve("hipMemGetInfo");
	real.hipMemPrefetchAsync =
		(fn_hipMemPrefetchAsync)resolve("hipMemPrefetchAsync");
	real.hipMemAdvise = (fn_hipMemAdvise)resolve("hipMemAdvise");
	real.hipDeviceSynchronize =
		(fn_hipDeviceSynchronize)resolve("hipDeviceSynchronize");
	real.hipSetDevice = (fn_hipSetDevice)resolve("hipSetDevice");
	real.hipGetDevice = (fn_hipGetDevice)resolve("hipGetDevice");
	real.hipStreamSynchronize =
		(fn_hipStreamSynchronize)resolve("hipStreamSynchronize");
	real.hipStreamCreateWithFlags = (fn_hipStreamCreateWithFlags)
		resolve("hipStreamCreateWithFlags");
	real.hipStreamDestroy =
		(fn_hipStreamDestroy)resolve("hipStreamDestroy");
	real.hipLaunchKernel = (fn_hipLaunchKernel)resolve("hipLaunchKernel");
	real.hipExtLaunchKernel =
		(fn_hipExtLaunchKernel)resolve("hipExtLaunchKernel");
	real.hipLaunchCooperativeKernel = (fn_hipLaunchCooperativeKernel)
		resolve("hipLaunchCooperativeKernel");
	real.hipModuleLaunchKernel =
		(fn_hipModuleLaunchKernel)resolve("hipModuleLaunchKernel");
	real.hipExtModuleLaunchKernel = (fn_hipExtModuleLaunchKernel)
		resolve("hipExtModuleLaunchKernel");
	real.hipGraphLaunch = (fn_hipGraphLaunch)resolve("hipGraphLaunch");
	real.hipMemcpy = (fn_hipMemcpy)resolve("hipMemcpy");
	real.hipMemcpyAsync = (fn_hipMemcpyAsync)resolve("hipMemcpyAsync");
	real.hipMemcpyWithStream =
		(fn_hipMemcpyWithStream)resolve("hipMemcpyWithStream");
	real.hipMemcpyHtoD = (fn_hipMemcpyHtoD)resolve("hipMemcpyHtoD");
	real.hipMemcpyDtoH = (fn_hipMemcpyDtoH)resolve("hipMemcpyDtoH");
	real.hipMemcpyDtoD = (fn_hipMemcpyDtoD)resolve("hipMemcpyDtoD");
	real.hipMemcpyHtoDAsync =
		(fn_hipMemcpyHtoDAsync)resolve("hipMemcpyHtoDAsync");
	real.hipMemcpyDtoHAsync =
		(fn_hipMemcpyDtoHAsync)resolve("hipMemcpyDtoHAsync");
	real.hipMemcpyDtoDAsync =
		(fn_hipMemcpyDtoDAsync)resolve("hipMemcpyDtoDAsync");
	real.hipMemset = (fn_hipMemset)resolve("hipMemset");
	real.hipMemsetAsync = (fn_hipMemsetAsync)resolve("hipMemsetAsync");
	real.hipMemsetD32Async =
		(fn_hipMemsetD32Async)resolve("hipMemsetD32Async");
	real.hipMemcpy2D = (fn_hipMemcpy2D)resolve("hipMemcpy2D");
	real.hipMemcpy2DAsync =
		(fn_hipMemcpy2DAsync)resolve("hipMemcpy2DAsync");
	real.hipMemcpyToSymbol =
		(fn_hipMemcpyToSymbol)resolve("hipMemcpyToSymbol");
	real.hipMemcpyFromSymbol =
		(fn_hipMemcpyFromSymbol)resolve("hipMemcpyFromSymbol");
	real.hipMemcpyPeerAsync =
		(fn_hipMemcpyPeerAsync)resolve("hipMemcpyPeerAsync");
	real.hipModuleLaunchCooperativeKernel =
		(fn_hipModuleLaunchCooperativeKernel)
		resolve("hipModuleLaunchCooperativeKernel");
	real.hipGetProcAddress =
		(fn_hipGetProcAddress)resolve("hipGetProcAddress");
	real.hipMemCreate = (fn_hipMemCreate)resolve("hipMemCreate");
	real.hipMemRelease = (fn_hipMemRelease)resolve("hipMemRelease");
	real.hipMemMap = (fn_hipMemMap)resolve("hipMemMap");
	real.hipMemUnmap = (fn_hipMemUnmap)resolve("hipMemUnmap");
	real.hipGetErrorString =
		(fn_hipGetErrorString)resolve("hipGetErrorString");

	if (real.hipMalloc == NULL || real.hipMallocManaged == NULL)
		log_fatal("hook: cannot resolve libamdhip64 entry points "
			  "(is the app a HIP/ROCm program?)");

	if (nvs_env_bool("NVSHARE_DEBUG", 0)) {
		Dl_info di;

		if (dladdr((void *)real.hipMalloc, &di) != 0)
			log_debug("hook: real hipMalloc = %p from %s",
				  (void *)real.hipMalloc,
				  di.dli_fname ? di.dli_fname : "?");
	}

	oversub_allowed = nvs_env_bool("NVSHARE_ENABLE_SINGLE_OVERSUB", 0);
	disable_um = nvs_env_bool("NVSHARE_DISABLE_UM", 0);
	alloc_prefetch = nvs_env_bool("NVSHARE_ALLOC_PREFETCH", 0);
	coarse_grain = nvs_env_bool("NVSHARE_COARSE_GRAIN", 1);
	preferred_loc = nvs_env_bool("NVSHARE_PREFERRED_LOC", 1);
	thp_advise = nvs_env_bool("NVSHARE_THP", 0);
	alloc_memset = nvs_env_bool("NVSHARE_ALLOC_MEMSET", 0);
	/* Reserve sized for 288 GB HBM3E; the reference used 1536 MiB on a
	 * 16 GB P100 (hook.c:45). */
	mem_reserve = (size_t)nvs_env_long("NVSHARE_RESERVE_MIB", 8192, 0,
					   1024 * 1024) * NVS_MIB;
	window = nvs_env_long("NVSHARE_WINDOW_START", 16, 1, 1 << 20);
	window_max = nvs_env_long("NVSHARE_WINDOW_MAX", 2048, 1, 1 << 20);
	sync_slow_ms = nvs_env_long("NVSHARE_SYNC_SLOW_MS", 1000, 1,
				    600000);
	sync_very_slow_ms = nvs_env_long("NVSHARE_SYNC_VERY_SLOW_MS", 10000,
					 1, 600000);

	{
		const char *x = getenv("HSA_XNACK");

		if (!disable_um && (x == NULL || x[0] != '1'))
			log_warn("HSA_XNACK is not '1': gfx950 demand "
				 "paging degrades to coarse-grained "
				 "migration; set HSA_XNACK=1 in the "
				 "client environment");
	}

	log_debug("hook: bootstrap done (oversub=%d disable_um=%d "
		  "reserve=%zu MiB)", oversub_allowed, disable_um,
		  mem_reserve / NVS_MIB);

	nvs_client_init();
}

#define BOOTSTRAP() pthread_once(&bootstrap_once, bootstrap)

#define CHECK_REAL(fn)                                                     \
	do {                                                               \
		if (real.fn == NULL)                                       \
			log_fatal("hook: real %s unavailable", #fn);       \
	} while (0)

static void memquery_init(void)
{
	size_t free_b = 0, total_b = 0;
	long fake;

	CHECK_REAL(hipMemGetInfo);
	if (real.hipMemGetInfo(&free_b, &total_b) != NVSHIP_SUCCESS)
		log_warn("hook: hipMemGetInfo failed; assuming 288 GiB");
	if (total_b == 0)
		total_b = 288ULL * 1024 * NVS_MIB;
	fake = nvs_env_long("NVSHARE_FAKE_TOTAL_MIB", 0, 0, 512L * 1024);
	if (fake > 0)
		total_b = (size_t)fake * NVS_MIB;
	mem_total = total_b;
	log_debug("hook: device memory total=%zu MiB reserve=%zu MiB",
		  mem_total / NVS_MIB, mem_reserve / NVS_MIB);
}

static size_t mem_limit(void)
{
	pthread_once(&memquery_once, memquery_init);
	return mem_total > mem_reserve ? mem_total - mem_reserve : 0;
}

/* ------------------------------------------------------------------ */
/* Allocation tracking                                                 */
/* ------------------------------------------------------------------ */

#define NVSHIP_MEM_ADVISE_SET_COARSE_GRAIN 100

/*
 * gfx950 managed-memory fast path.
 *
 * Every converted allocation is advised (a) coarse-grain (whole-range
 * migration granularity, full-rate access, working hardware FP
 * atomics; fine-grain host/device *concurrent* access is not
 * something the hipMalloc contract we replace ever promised) and (b)
 * preferred-location = the app's device, so first-touch pages are
 * born in HBM instead of migrating later.  Both advise calls are
 * metadata-only: no data moves, nothing can race the app's writes.
 *
 * An earlier design ALSO issued an eager hipMemPrefetchAsync here
 * (the round-1 NVSHARE_ALLOC_PREFETCH=1 default).  Measured on ROCm
 * 7.2 / MI355X that CORRUPTS the allocation: the null-stream prefetch
 * races PyTorch's initialization writes on non-blocking streams and
 * loses some of them — ResNet-50 weights came up garbage (initial
 * loss 930 vs 7.17) or trained to NaN within 3 steps, with bad grads
 * in every layer (profiles/nanhunt.log, tools/nanhunt.{py,sh}: every
 * arm with alloc-prefetch on corrupts, every arm without it matches
 * stock exactly).  The env survives for experiments but defaults OFF.
 * Returns 1 when populated now (0 = deferred to the gate).
 */
static int populate_managed(void *ptr, size_t size)
{
	if (thp_advise)
		madvise(ptr, size, MADV_HUGEPAGE);
	if (coarse_grain && real.hipMemAdvise != NULL)
		real.hipMemAdvise(ptr, size,
				  NVSHIP_MEM_ADVISE_SET_COARSE_GRAIN, 0);
	if (preferred_loc && real.hipMemAdvise != NULL)
		real.hipMemAdvise(ptr, size,
				  NVSHIP_MEM_ADVISE_SET_PREFERRED_LOCATION,
				  nvs_app_device);
	if (alloc_memset && real.hipMemset != NULL &&
	    real.hipDeviceSynchronize != NULL) {
		/* Device-side first touch: the fill kernel's writes
		 * materialize the pages in HBM (preferred location),
		 * and the sync completes before the app ever sees the
		 * pointer — safe by construction, no prefetch
		 * machinery involved. */
		real.hipMemset(ptr, 0, size);
		real.hipDeviceSynchronize();
		return 1;
	}
	if (!alloc_prefetch || real.hipMemPrefetchAsync == NULL)
		return 1; /* nothing to defer */
	if (!nvs_can_submit_now())
		return 0; /* defer to the gate */
	real.hipMemPrefetchAsync(ptr, size, nvs_app_device, NULL);
	return 1;
}

/* reserved=1: the size was already added to sum_allocated by the
 * caller (cap reservation under alloc_mutex before the real alloc —
 * check-then-act otherwise lets concurrent allocations jointly
 * exceed the cap). */
static void track_alloc2(void *ptr, size_t size, int populated,
			 int reserved, int passthrough)
{
	struct nvs_alloc *a = malloc(sizeof(*a));

	if (a == NULL)
		return;
	a->ptr = ptr;
	a->size = size;
	a->populated = populated;
	a->bulk_ready = 0;
	a->passthrough = passthrough;
	pthread_mutex_lock(&alloc_mutex);
	a->next = alloc_list;
	alloc_list = a;
	if (!reserved)
		sum_allocated += size;
	if (!populated)
		__atomic_fetch_add(&n_unpopulated, 1, __ATOMIC_RELAXED);
	pthread_mutex_unlock(&alloc_mutex);
	log_debug("hook: +alloc %p %zu MiB (sum %zu MiB%s%s)", ptr,
		  size / NVS_MIB, sum_allocated / NVS_MIB,
		  populated ? "" : ", population deferred",
		  passthrough ? ", passthrough" : "");
}

static void track_alloc(void *ptr, size_t size, int populated,
			int reserved)
{
	track_alloc2(ptr, size, populated, reserved, 0);
}

/* Reserve size against the cap (sum_allocated) before allocating.
 * Returns 0 on success, -1 when the cap would be exceeded. */
static int reserve_cap(size_t size)
{
	pthread_mutex_lock(&alloc_mutex);
	/* Held-but-freed (cached) ranges still occupy memory and count
	 * here too; the caller flushes the cache and retries when this
	 * rejects (malloc_managed). */
	if (sum_allocated + free_cache_bytes + size > mem_limit()) {
		size_t sum = sum_allocated;

		pthread_mutex_unlock(&alloc_mutex);
		log_debug("hook: reject alloc of %zu MiB (sum %zu MiB, "
			  "limit %zu MiB); set "
			  "NVSHARE_ENABLE_SINGLE_OVERSUB=1 to oversubscribe",
			  size / NVS_MIB, sum / NVS_MIB,
			  mem_limit() / NVS_MIB);
		return -1;
	}
	sum_allocated += size;
	pthread_mutex_unlock(&alloc_mutex);
	return 0;
}

static void unreserve_cap(size_t size)
{
	pthread_mutex_lock(&alloc_mutex);
	sum_allocated -= size;
	pthread_mutex_unlock(&alloc_mutex);
}

long nvs_sum_allocated_mib(void)
{
	size_t n;

	pthread_mutex_lock(&alloc_mutex);
	n = sum_allocated;
	pthread_mutex_unlock(&alloc_mutex);
	return (long)(n / NVS_MIB);
}

/* Advertised device capacity (MiB), for MEM_UPDATE so the scheduler
 * can judge whole-node memory pressure. */
long nvs_mem_total_mib(void)
{
	(void)mem_limit(); /* ensure memquery ran */
	return (long)(mem_total / NVS_MIB);
}

/* Called from the gate with the submission read lock held and the GPU
 * lock owned: materialize any deferred ranges before real work. */
void nvs_populate_pending(void)
{
	struct nvs_alloc *a;
	int any = 0;

	if (__atomic_load_n(&n_unpopulated, __ATOMIC_RELAXED) == 0)
		return;
	if (real.hipMemPrefetchAsync == NULL)
		return;
	pthread_mutex_lock(&alloc_mutex);
	for (a = alloc_list; a != NULL; a = a->next) {
		if (a->populated)
			continue;
		real.hipMemPrefetchAsync(a->ptr, a->size, nvs_app_device,
					 NULL);
		a->populated = 1;
		__atomic_fetch_sub(&n_unpopulated, 1, __ATOMIC_RELAXED);
		any = 1;
	}
	pthread_mutex_unlock(&alloc_mutex);
	if (any)
		log_debug("hook: populated deferred ranges");
}

/* Returns tracked size, or 0 if unknown pointer.  was_passthrough
 * (optional) reports whether the range was a real-VRAM passthrough. */
static size_t untrack_alloc2(void *ptr, int *was_passthrough)
{
	struct nvs_alloc **pp, *a;
	size_t size = 0;

	if (was_passthrough != NULL)
		*was_passthrough = 0;
	pthread_mutex_lock(&alloc_mutex);
	for (pp = &alloc_list; *pp != NULL; pp = &(*pp)->next) {
		if ((*pp)->ptr == ptr) {
			a = *pp;
			*pp = a->next;
			size = a->size;
			sum_allocated -= size;
			if (was_passthrough != NULL)
				*was_passthrough = a->passthrough;
			if (!a->populated)
				__atomic_fetch_sub(&n_unpopulated, 1,
						   __ATOMIC_RELAXED);
			free(a);
			break;
		}
	}
	pthread_mutex_unlock(&alloc_mutex);
	if (size != 0)
		log_debug("hook: -alloc %p %zu MiB (sum %zu MiB)", ptr,
			  size / NVS_MIB, sum_allocated / NVS_MIB);
	return size;
}

/*
 * Managed-range free cache.  Measured (profiles/ab5 census): the
 * torch/MIOpen stack allocates and frees one small (2-128 MiB)
 * segment per training step; every fresh hipMallocManaged range
 * first-touch faults at page speed, which is exactly the episodic
 * 100-215 ms kernel-call tail that costs the ~1.4x managed step
 * overhead (profiles/RESULTS.md §14).  Freed managed ranges are
 * therefore kept (pages stay resident) and recycled on the next
 * same-size allocation.  Bounded by NVSHARE_FREE_CACHE_MIB (default
 * 1024); flushed when the cap would reject an allocation and before
 * pressure eviction.  Note hipFree's implicit device synchronization
 * is skipped for cached frees — the same contract relaxation the
 * managed conversion already makes.
 */
struct nvs_cached {
	void *ptr;
	size_t size;
	struct nvs_cached *next;
};
static struct nvs_cached *free_cache;

static size_t free_cache_cap(void)
{
	static long mib = -1;

	if (mib < 0)
		mib = nvs_env_long("NVSHARE_FREE_CACHE_MIB", 1024, 0,
				   1024 * 1024);
	return (size_t)mib * NVS_MIB;
}

/* fork() child: drop the cache VIEW without freeing — the ranges
 * belong to the parent's GPU context; recycling them in the child
 * would alias live parent memory (client.c atfork_child). */
void nvs_free_cache_forget(void)
{
	struct nvs_cached *c;

	while (free_cache != NULL) {
		c = free_cache;
		free_cache = c->next;
		free(c);
	}
	free_cache_bytes = 0;
}

/* Caller must NOT hold alloc_mutex. */
static void flush_free_cache(void)
{
	struct nvs_cached *list, *c;

	pthread_mutex_lock(&alloc_mutex);
	list = free_cache;
	free_cache = NULL;
	free_cache_bytes = 0;
	pthread_mutex_unlock(&alloc_mutex);
	while (list != NULL) {
		c = list;
		list = c->next;
		if (real.hipFree != NULL)
			real.hipFree(c->ptr);
		free(c);
	}
}

/* Try to serve an allocation from the cache (exact size match; the
 * per-step churn repeats identical sizes).  Returns ptr or NULL. */
static void *free_cache_pop(size_t size)
{
	struct nvs_cached **pp, *c;
	void *ptr = NULL;

	pthread_mutex_lock(&alloc_mutex);
	for (pp = &free_cache; *pp != NULL; pp = &(*pp)->next) {
		if ((*pp)->size == size) {
			c = *pp;
			*pp = c->next;
			ptr = c->ptr;
			free_cache_bytes -= size;
			free(c);
			break;
		}
	}
	pthread_mutex_unlock(&alloc_mutex);
	return ptr;
}

/* Stash a managed range instead of freeing it.  Returns 1 when
 * cached (caller must NOT call the real free). */
static int free_cache_push(void *ptr, size_t size)
{
	struct nvs_cached *c;

	if (size == 0 || size > free_cache_cap())
		return 0;
	c = malloc(sizeof(*c));
	if (c == NULL)
		return 0;
	c->ptr = ptr;
	c->size = size;
	pthread_mutex_lock(&alloc_mutex);
	if (size + free_cache_bytes > free_cache_cap()) {
		pthread_mutex_unlock(&alloc_mutex);
		free(c);
		return 0;
	}
	c->next = free_cache;
	free_cache = c;
	free_cache_bytes += size;
	pthread_mutex_unlock(&alloc_mutex);
	log_debug("hook: cached free %p %zu MiB (cache %zu MiB)", ptr,
		  size / NVS_MIB, free_cache_bytes / NVS_MIB);
	return 1;
}

/*
 * Prefetch tracked allocations back to the device after a lock handoff
 * (called from the client thread on LOCK_OK when NVSHARE_PREFETCH=1).
 * Restore bandwidth is HMM-migration-bound (profiles/restorebench.json:
 * 1.9 GB/s on one stream, 3.8 GB/s with 4 streams x 1 GiB chunks on
 * MI355X), so the chunks are spread round-robin over 4 dedicated
 * non-blocking streams.  Fire-and-forget: pages the app touches before
 * their chunk lands just retry-fault as usual.
 */
#define PREFETCH_STREAMS 4
#define PREFETCH_CHUNK (1024ULL * NVS_MIB)
static nvship_stream_t prefetch_streams[PREFETCH_STREAMS];
static int prefetch_streams_ready;

static void ensure_prefetch_streams(void)
{
	int i, ok = 1;

	if (prefetch_streams_ready)
		return;
	for (i = 0; i < PREFETCH_STREAMS; i++) {
		if (real.hipStreamCreateWithFlags == NULL ||
		    real.hipStreamCreateWithFlags(
			&prefetch_streams[i],
			NVSHIP_STREAM_NON_BLOCKING) != NVSHIP_SUCCESS) {
			ok = 0;
			break;
		}
	}
	if (!ok)
		for (i = 0; i < PREFETCH_STREAMS; i++)
			prefetch_streams[i] = NULL;
	prefetch_streams_ready = 1;
}

static void sync_prefetch_streams(void)
{
	int i;

	if (real.hipStreamSynchronize == NULL)
		return;
	for (i = 0; i < PREFETCH_STREAMS; i++)
		real.hipStreamSynchronize(prefetch_streams[i]);
}

/* Real (driver) free memory — NOT the advertised lie. */
static size_t real_free_bytes(void)
{
	size_t freeb = 0, totalb = 0;

	if (real.hipMemGetInfo == NULL ||
	    real.hipMemGetInfo(&freeb, &totalb) != NVSHIP_SUCCESS)
		return (size_t)-1;
	return freeb;
}

void nvs_prefetch_allocs(void)
{
	struct nvs_alloc *a;
	size_t budget;
	int s = 0;
	int64_t t0 = nvs_now_ns();

	if (real.hipMemPrefetchAsync == NULL)
		return;
	ensure_prefetch_streams();
	budget = (size_t)nvs_env_long("NVSHARE_PREFETCH_MIB", 0, 0,
				      1024 * 1024) * NVS_MIB;
	if (budget == 0)
		budget = (size_t)-1; /* whole tracked set */
	pthread_mutex_lock(&alloc_mutex);
	/* Newest-first: the list is LIFO, which approximates MRU. */
	for (a = alloc_list; a != NULL && budget > 0; a = a->next) {
		size_t left = a->size < budget ? a->size : budget;
		char *p = a->ptr;

		if (a->passthrough)
			continue; /* real VRAM: never migrated */
		if (!a->populated) {
			a->populated = 1;
			__atomic_fetch_sub(&n_unpopulated, 1,
					   __ATOMIC_RELAXED);
		}

		while (left > 0) {
			size_t n = left < PREFETCH_CHUNK ? left :
				   PREFETCH_CHUNK;

			real.hipMemPrefetchAsync(p, n, nvs_app_device,
						 prefetch_streams[s]);
			s = (s + 1) % PREFETCH_STREAMS;
			p += n;
			budget -= n;
			left -= n;
		}
	}
	pthread_mutex_unlock(&alloc_mutex);
	/* Blocking: the gate opens only after the working set is back,
	 * otherwise the app's demand faults (0.15 GB/s) race the bulk
	 * migration (3.8 GB/s) and lose. */
	sync_prefetch_streams();
	log_debug("hook: restore prefetch took %lld ms",
		  (long long)((nvs_now_ns() - t0) / 1000000));
}

/*
 * Evict tracked allocations to host after giving up the lock
 * (called from client.c BEFORE LOCK_RELEASED; on by default via
 * NVSHARE_AUTO_MIGRATE, self-gated on real memory pressure below).
 * Explicit eviction runs at ~10.9 GB/s on MI355X vs ~0.15 GB/s when
 * the next client's demand faults push pages out one at a time
 * (profiles/restorebench.json).  Blocking: the next client is granted
 * only once the room actually exists, otherwise its restore races the
 * driver's fault-driven eviction of our pages.
 */
void nvs_evict_allocs(void)
{
	struct nvs_alloc *a;
	int s = 0;
	size_t resident, freeb;
	int64_t t0 = nvs_now_ns();

	/* Held-but-freed ranges are dead weight under pressure: free
	 * them outright instead of migrating them. */
	flush_free_cache();
	if (real.hipMemPrefetchAsync == NULL)
		return;
	ensure_prefetch_streams();
	/* Only evict under actual memory pressure: when the device has
	 * room for another client's set alongside ours, eviction would
	 * just churn migrations. */
	pthread_mutex_lock(&alloc_mutex);
	resident = sum_allocated;
	pthread_mutex_unlock(&alloc_mutex);
	freeb = real_free_bytes();
	if (freeb != (size_t)-1 && freeb > resident) {
		log_debug("hook: skip evict (free %zu MiB > set %zu MiB)",
			  freeb / NVS_MIB, resident / NVS_MIB);
		return;
	}
	pthread_mutex_lock(&alloc_mutex);
	for (a = alloc_list; a != NULL; a = a->next) {
		char *p = a->ptr;
		size_t left = a->size;

		if (a->passthrough)
			continue; /* real VRAM: never migrated */
		while (left > 0) {
			size_t n = left < PREFETCH_CHUNK ? left :
				   PREFETCH_CHUNK;

			real.hipMemPrefetchAsync(p, n,
						 NVSHIP_CPU_DEVICE_ID,
						 prefetch_streams[s]);
			s = (s + 1) % PREFETCH_STREAMS;
			p += n;
			left -= n;
		}
	}
	pthread_mutex_unlock(&alloc_mutex);
	/* Blocking: LOCK_RELEASED is sent after the room exists. */
	sync_prefetch_streams();
	log_debug("hook: evict to host took %lld ms",
		  (long long)((nvs_now_ns() - t0) / 1000000));
}

/* ------------------------------------------------------------------ */
/* Pending-kernel window                                               */
/* ------------------------------------------------------------------ */

int nvs_scheduler_gating(void); /* from client.c (racy read is fine) */

static void after_launch(void)
{
	long n;
	int64_t t0, dt_ms;

	if (!nvs_scheduler_gating())
		return;
	pthread_mutex_lock(&win_mutex);
	n = ++kern_since_sync;
	if (n < window) {
		pthread_mutex_unlock(&win_mutex);
		return;
	}
	kern_since_sync = 0;
	pthread_mutex_unlock(&win_mutex);

	/* If the lock was just lost, the DROP_LOCK drain will sync; a
	 * sync here would block on the NEXT holder's kernels and poison
	 * the window adaptation with their runtime. */
	if (!nvs_can_submit_now())
		return;
	CHECK_REAL(hipDeviceSynchronize);
	t0 = nvs_now_ns();
	real.hipDeviceSynchronize();
	dt_ms = (nvs_now_ns() - t0) / 1000000;

	pthread_mutex_lock(&win_mutex);
	if (dt_ms >= sync_very_slow_ms)
		window = 1;
	else if (dt_ms >= sync_slow_ms)
		window = window / 2 > 1 ? window / 2 : 1;
	else
		window = window * 2 < window_max ? window * 2 : window_max;
	pthread_mutex_unlock(&win_mutex);
	log_debug("hook: window sync %lld ms -> window=%ld",
		  (long long)dt_ms, window);
}

/* ------------------------------------------------------------------ */
/* Hook call counters (rocprofv3 cross-check of interposition          */
/* completeness, the thesis Table 11.6 methodology: our per-symbol     */
/* counts must equal the profiler's API counts).  Dumped at exit when  */
/* NVSHARE_DEBUG is set.                                               */
/* ------------------------------------------------------------------ */

enum hook_id {
	H_hipMalloc, H_hipMallocManaged, H_hipExtMallocWithFlags,
	H_hipMallocAsync, H_hipMallocFromPoolAsync, H_hipFree,
	H_hipFreeAsync, H_hipMemGetInfo, H_hipSetDevice,
	H_hipLaunchKernel, H_hipExtLaunchKernel,
	H_hipLaunchCooperativeKernel, H_hipModuleLaunchKernel,
	H_hipExtModuleLaunchKernel, H_hipModuleLaunchCooperativeKernel,
	H_hipGraphLaunch, H_hipMemcpy, H_hipMemcpyAsync,
	H_hipMemcpyWithStream, H_hipMemcpyHtoD, H_hipMemcpyDtoH,
	H_hipMemcpyDtoD, H_hipMemcpyHtoDAsync, H_hipMemcpyDtoHAsync,
	H_hipMemcpyDtoDAsync, H_hipMemcpy2D, H_hipMemcpy2DAsync,
	H_hipMemcpyToSymbol, H_hipMemcpyFromSymbol, H_hipMemcpyPeerAsync,
	H_hipMemset, H_hipMemsetAsync, H_hipMemsetD32Async,
	H_hipGetProcAddress, H_dlsym,
	H_hipMemCreate, H_hipMemRelease, H_hipMemMap,
	H_COUNT_
};

static const char *hook_names[H_COUNT_] = {
	"hipMalloc", "hipMallocManaged", "hipExtMallocWithFlags",
	"hipMallocAsync", "hipMallocFromPoolAsync", "hipFree",
	"hipFreeAsync", "hipMemGetInfo", "hipSetDevice",
	"hipLaunchKernel", "hipExtLaunchKernel",
	"hipLaunchCooperativeKernel", "hipModuleLaunchKernel",
	"hipExtModuleLaunchKernel", "hipModuleLaunchCooperativeKernel",
	"hipGraphLaunch", "hipMemcpy", "hipMemcpyAsync",
	"hipMemcpyWithStream", "hipMemcpyHtoD", "hipMemcpyDtoH",
	"hipMemcpyDtoD", "hipMemcpyHtoDAsync", "hipMemcpyDtoHAsync",
	"hipMemcpyDtoDAsync", "hipMemcpy2D", "hipMemcpy2DAsync",
	"hipMemcpyToSymbol", "hipMemcpyFromSymbol", "hipMemcpyPeerAsync",
	"hipMemset", "hipMemsetAsync", "hipMemsetD32Async",
	"hipGetProcAddress", "dlsym",
	"hipMemCreate", "hipMemRelease", "hipMemMap",
};

static unsigned long hook_counts[H_COUNT_];
static unsigned long hook_ns[H_COUNT_];     /* NVSHARE_PROFILE_HOOKS */
static int profile_hooks = -1;

#define BUMP(id) __atomic_fetch_add(&hook_counts[id], 1UL, \
				    __ATOMIC_RELAXED)

/* Wall-time accounting per hook (sum of wrapper time incl. the real
 * call): `NVSHARE_PROFILE_HOOKS=1 NVSHARE_DEBUG=1` dumps per-hook
 * total ms at exit — the tool for attributing interposer overhead
 * (docs/roadmap.md #1). */
static inline int64_t prof_begin(void)
{
	if (profile_hooks < 0)
		profile_hooks = nvs_env_bool("NVSHARE_PROFILE_HOOKS", 0);
	return profile_hooks ? nvs_now_ns() : 0;
}

static inline void prof_end(int id, int64_t t0)
{
	if (t0 != 0)
		__atomic_fetch_add(&hook_ns[id],
				   (unsigned long)(nvs_now_ns() - t0),
				   __ATOMIC_RELAXED);
}

__attribute__((destructor)) static void dump_hook_counts(void)
{
	int i, any = 0;
	char line[2048];
	size_t off = 0;

	if (!nvs_debug_enabled)
		return;
	for (i = 0; i < H_COUNT_; i++) {
		if (hook_counts[i] == 0)
			continue;
		any = 1;
		off += (size_t)snprintf(line + off, sizeof(line) - off,
					"%s%s=%lu", off ? " " : "",
					hook_names[i], hook_counts[i]);
		if (off >= sizeof(line) - 64)
			break;
	}
	if (any)
		log_debug("hook call counts: %s", line);
	if (profile_hooks == 1) {
		off = 0;
		line[0] = '\0';
		for (i = 0; i < H_COUNT_; i++) {
			if (hook_ns[i] == 0)
				continue;
			off += (size_t)snprintf(line + off,
						sizeof(line) - off,
						"%s%s=%.1fms",
						off ? " " : "",
						hook_names[i],
						hook_ns[i] / 1e6);
			if (off >= sizeof(line) - 64)
				break;
		}
		log_debug("hook wall time: %s", line);
	}
}

/* ------------------------------------------------------------------ */
/* Hooked entry points                                                 */
/* ------------------------------------------------------------------ */

/* Shared managed-conversion path.  Counters are bumped only at the
 * public entry points so delegating wrappers don't double-count. */
static nvshipError_t malloc_managed(void **ptr, size_t size)
{
	nvshipError_t r;

	CHECK_REAL(hipMalloc);
	if (ptr == NULL)
		return NVSHIP_ERROR_INVALID_VALUE;
	if (size == 0)
		return real.hipMalloc(ptr, size);
	if (disable_um)
		return real.hipMalloc(ptr, size);
	if (!oversub_allowed && reserve_cap(size) != 0) {
		/* Cached (held-but-freed) ranges may be what's in the
		 * way: release them and retry once. */
		flush_free_cache();
		if (reserve_cap(size) != 0)
			return NVSHIP_ERROR_OOM;
	}
	{
		void *cached = free_cache_pop(size);

		if (cached != NULL) {
			/* Recycled range: pages already resident and
			 * advised — no first-touch fault storm. */
			*ptr = cached;
			track_alloc(cached, size, 1, !oversub_allowed);
			return NVSHIP_SUCCESS;
		}
	}
	r = real.hipMallocManaged(ptr, size, NVSHIP_MEM_ATTACH_GLOBAL);
	if (r == NVSHIP_SUCCESS)
		track_alloc(*ptr, size, populate_managed(*ptr, size),
			    !oversub_allowed);
	else if (!oversub_allowed)
		unreserve_cap(size);
	return r;
}

nvshipError_t hipMalloc(void **ptr, size_t size)
{
	BOOTSTRAP();
	BUMP(H_hipMalloc);
	return malloc_managed(ptr, size);
}

nvshipError_t hipExtMallocWithFlags(void **ptr, size_t size,
				    unsigned int flags)
{
	BOOTSTRAP();
	BUMP(H_hipExtMallocWithFlags);
	if (disable_um && real.hipExtMallocWithFlags != NULL)
		return real.hipExtMallocWithFlags(ptr, size, flags);
	/* All flag variants become managed; the flags are advisory. */
	(void)flags;
	return malloc_managed(ptr, size);
}

/*
 * Stream-ordered allocations: PyTorch/MIOpen allocate a transient
 * workspace this way EVERY STEP (measured: 60 hipMallocFromPoolAsync +
 * 60 frees over 60 ResNet steps).  Converting those to managed costs
 * a populate plus a blocking stream-sync on free per step — the
 * measured ~1.27x ResNet overhead.  Small stream-ordered allocations
 * are therefore passed through to the real allocator (transient,
 * never a shareable working set); only large ones
 * (> NVSHARE_PASSTHROUGH_MIB, default 256) are converted and tracked.
 */
static int pool_passthrough(size_t size)
{
	static long thresh_mib = -1;

	if (thresh_mib < 0)
		thresh_mib = nvs_env_long("NVSHARE_PASSTHROUGH_MIB", 256,
					  0, 1024 * 1024);
	return !disable_um && size <= (size_t)thresh_mib * NVS_MIB;
}

/* Passthrough allocations stay real VRAM but are still COUNTED: they
 * reserve against the cap, appear in hipMemGetInfo accounting and in
 * MEM_UPDATE, and are untracked on free (the reference counts every
 * byte, hook.c:662-670; a client must not be able to dodge the cap by
 * allocating its whole set in small pool chunks). */
static nvshipError_t pool_alloc_passthrough(
	void **ptr, size_t size,
	nvshipError_t (*do_alloc)(void **, size_t))
{
	nvshipError_t r;

	if (!disable_um && !oversub_allowed && size > 0 &&
	    reserve_cap(size) != 0)
		return NVSHIP_ERROR_OOM;
	r = do_alloc(ptr, size);
	if (r == NVSHIP_SUCCESS) {
		if (!disable_um && size > 0)
			track_alloc2(*ptr, size, 1, !oversub_allowed, 1);
	} else if (!disable_um && !oversub_allowed && size > 0) {
		unreserve_cap(size);
	}
	return r;
}

/* Thunks adapting stream/pool args for pool_alloc_passthrough. */
static __thread nvship_stream_t pp_stream;
static __thread nvship_mempool_t pp_pool;

static nvshipError_t do_malloc_async(void **ptr, size_t size)
{
	return real.hipMallocAsync(ptr, size, pp_stream);
}

static nvshipError_t do_malloc_from_pool_async(void **ptr, size_t size)
{
	return real.hipMallocFromPoolAsync(ptr, size, pp_pool, pp_stream);
}

nvshipError_t hipMallocAsync(void **ptr, size_t size, nvship_stream_t s)
{
	BOOTSTRAP();
	BUMP(H_hipMallocAsync);
	if (disable_um && real.hipMallocAsync != NULL)
		return real.hipMallocAsync(ptr, size, s);
	if (pool_passthrough(size) && real.hipMallocAsync != NULL) {
		pp_stream = s;
		return pool_alloc_passthrough(ptr, size, do_malloc_async);
	}
	/* Large stream-ordered alloc becomes an immediate managed alloc:
	 * the pointer is valid earlier than required (safe). */
	return malloc_managed(ptr, size);
}

nvshipError_t hipMallocFromPoolAsync(void **ptr, size_t size,
				     nvship_mempool_t pool,
				     nvship_stream_t s)
{
	BOOTSTRAP();
	BUMP(H_hipMallocFromPoolAsync);
	if (disable_um && real.hipMallocFromPoolAsync != NULL)
		return real.hipMallocFromPoolAsync(ptr, size, pool, s);
	if (pool_passthrough(size) && real.hipMallocFromPoolAsync != NULL) {
		pp_stream = s;
		pp_pool = pool;
		return pool_alloc_passthrough(ptr, size,
					      do_malloc_from_pool_async);
	}
	return malloc_managed(ptr, size);
}

nvshipError_t hipFree(void *ptr)
{
	BOOTSTRAP();
	BUMP(H_hipFree);
	CHECK_REAL(hipFree);
	if (ptr != NULL) {
		int was_pt = 0;
		size_t sz = untrack_alloc2(ptr, &was_pt);

		if (sz != 0 && !was_pt && !disable_um &&
		    free_cache_push(ptr, sz))
			return NVSHIP_SUCCESS;
	}
	return real.hipFree(ptr);
}

nvshipError_t hipFreeAsync(void *ptr, nvship_stream_t stream)
{
	int was_pt = 0;

	BOOTSTRAP();
	BUMP(H_hipFreeAsync);
	if (ptr == NULL)
		return NVSHIP_SUCCESS;
	{
		size_t sz = untrack_alloc2(ptr, &was_pt);

		if (sz != 0 && !was_pt) {
			/* We converted this to a managed alloc:
			 * stream-ordered free semantics require prior
			 * stream work to finish before the range can
			 * be reused or released. */
			CHECK_REAL(hipFree);
			if (real.hipStreamSynchronize != NULL)
				real.hipStreamSynchronize(stream);
			if (!disable_um && free_cache_push(ptr, sz))
				return NVSHIP_SUCCESS;
			return real.hipFree(ptr);
		}
	}
	CHECK_REAL(hipFreeAsync);
	return real.hipFreeAsync(ptr, stream);
}

nvshipError_t hipMemGetInfo(size_t *free_p, size_t *total_p)
{
	size_t limit, freeb;

	BOOTSTRAP();
	BUMP(H_hipMemGetInfo);
	CHECK_REAL(hipMemGetInfo);
	if (disable_um)
		return real.hipMemGetInfo(free_p, total_p);
	limit = mem_limit();
	pthread_mutex_lock(&alloc_mutex);
	{
		/* Held-but-freed (cached) ranges still occupy memory:
		 * every byte counted (reference hook.c:662-670). */
		size_t used = sum_allocated + free_cache_bytes;

		freeb = used < limit ? limit - used : 0;
	}
	pthread_mutex_unlock(&alloc_mutex);
	if (free_p != NULL)
		*free_p = freeb;
	if (total_p != NULL)
		*total_p = mem_total;
	return NVSHIP_SUCCESS;
}

nvshipError_t hipSetDevice(int dev)
{
	BOOTSTRAP();
	BUMP(H_hipSetDevice);
	CHECK_REAL(hipSetDevice);
	nvs_app_device = dev;
	return real.hipSetDevice(dev);
}

/* ---- gated work submissions ---- */

/*
 * First-bulk-write residency for managed ranges.  An HtoD copy (or
 * large memset) into FRESH managed memory runs at page-fault speed
 * (~0.1 GB/s: ~17 ms per model-weight copy; ~10 s of a 25 s ResNet
 * job was memcpy wrappers at init — profiles/ab_r2.log hk_prof arm).
 *
 * SAFETY (measured, ROCm 7.2 gfx950): hipMemPrefetchAsync's
 * migration is NOT ordered with subsequent work enqueued on the same
 * stream — a prefetch issued right before the copy corrupts the data
 * just like the alloc-time variant did (gpurun_out/ab2.log: every
 * arm with the naive same-stream prefetch trained to NaN at 10x
 * slowdown).  The only safe pattern is prefetch + HOST SYNC before
 * any dependent write — exactly what the LOCK_OK restore path does.
 * So: on the FIRST large write into a tracked range, prefetch the
 * WHOLE range, hipStreamSynchronize, and mark it; later writes into
 * that range take the flag fast path.  One sync per allocation
 * amortizes over the many per-tensor uploads a model load does into
 * the same caching-allocator segment.  NVSHARE_COPY_PREFETCH=0
 * disables.
 */
static void copy_prefetch(void *dst, size_t n, nvship_stream_t s)
{
	static int enabled = -1;
	static long min_mib = 8;
	static pthread_mutex_t prep_mutex = PTHREAD_MUTEX_INITIALIZER;
	struct nvs_alloc *a;
	char *base = NULL, *d = dst;
	size_t size = 0;

	if (enabled < 0) {
		/* Default OFF: even the quiesce+prefetch+host-sync
		 * pattern corrupts when the range carries coarse-grain
		 * or preferred-location advise (gpurun_out/ab3.log:
		 * hk_plain clean, hooked/hk_nocg NaN) — on ROCm 7.2
		 * hipMemPrefetchAsync of never-touched advised ranges
		 * is not safe in-band at all.  Kept as an experiment
		 * knob for future driver versions. */
		enabled = nvs_env_bool("NVSHARE_COPY_PREFETCH", 0);
		min_mib = nvs_env_long("NVSHARE_COPY_PREFETCH_MIB", 8, 1,
				       1024 * 1024);
	}
	if (!enabled || disable_um || real.hipMemPrefetchAsync == NULL ||
	    real.hipStreamSynchronize == NULL)
		return;
	if (n < (size_t)min_mib * NVS_MIB)
		return;
	/* Serialized end to end: a second large writer into the same
	 * range must wait here until the first writer's migration has
	 * fully completed, not just until the flag flips. */
	pthread_mutex_lock(&prep_mutex);
	pthread_mutex_lock(&alloc_mutex);
	for (a = alloc_list; a != NULL; a = a->next) {
		char *p = a->ptr;

		if (!a->passthrough && d >= p && d < p + a->size) {
			if (!a->bulk_ready) {
				a->bulk_ready = 1;
				base = p;
				size = a->size;
			}
			break;
		}
	}
	pthread_mutex_unlock(&alloc_mutex);
	if (base == NULL) {
		pthread_mutex_unlock(&prep_mutex);
		return;
	}
	/* Quiesce the device first: an in-flight write into the same
	 * range from ANOTHER stream would race the migration (the
	 * documented-unsafe overlap).  Once per range, so the full
	 * drain amortizes. */
	if (real.hipDeviceSynchronize != NULL)
		real.hipDeviceSynchronize();
	real.hipMemPrefetchAsync(base, size, nvs_app_device, s);
	real.hipStreamSynchronize(s); /* migration MUST complete before
				       * the write lands (see above) */
	pthread_mutex_unlock(&prep_mutex);
}

#define GATED2(id, call)                                                   \
	do {                                                               \
		nvshipError_t r_;                                          \
		int64_t t0_ = prof_begin();                                \
		nvs_submit_begin();                                        \
		r_ = (call);                                               \
		nvs_submit_end();                                          \
		prof_end(id, t0_);                                         \
		return r_;                                                 \
	} while (0)

/* Gated submission that first prefetches the written range (dst) to
 * the device on the same stream. */
#define GATED_COPY(id, dst, n, stream, call)                               \
	do {                                                               \
		nvshipError_t r_;                                          \
		int64_t t0_ = prof_begin();                                \
		nvs_submit_begin();                                        \
		copy_prefetch((void *)(dst), (n), (stream));               \
		r_ = (call);                                               \
		nvs_submit_end();                                          \
		prof_end(id, t0_);                                         \
		return r_;                                                 \
	} while (0)

#define GATED_LAUNCH2(id, call)                                            \
	do {                                                               \
		nvshipError_t r_;                                          \
		int64_t t0_ = prof_begin();                                \
		nvs_submit_begin();                                        \
		r_ = (call);                                               \
		nvs_submit_end();                                          \
		after_launch();                                            \
		prof_end(id, t0_);                                         \
		return r_;                                                 \
	} while (0)

nvshipError_t hipLaunchKernel(const void *f, nvship_dim3 grid,
			      nvship_dim3 block, void **args,
			      size_t shmem, nvship_stream_t stream)
{
	BOOTSTRAP();
	BUMP(H_hipLaunchKernel);
	CHECK_REAL(hipLaunchKernel);
	GATED_LAUNCH2(H_hipLaunchKernel, real.hipLaunchKernel(f, grid, block, args, shmem,
					  stream));
}

nvshipError_t hipExtLaunchKernel(const void *f, nvship_dim3 grid,
				 nvship_dim3 block, void **args,
				 size_t shmem, nvship_stream_t stream,
				 nvship_event_t ev0, nvship_event_t ev1,
				 int flags)
{
	BOOTSTRAP();
	BUMP(H_hipExtLaunchKernel);
	CHECK_REAL(hipExtLaunchKernel);
	GATED_LAUNCH2(H_hipExtLaunchKernel, real.hipExtLaunchKernel(f, grid, block, args, shmem,
					     stream, ev0, ev1, flags));
}

nvshipError_t hipLaunchCooperativeKernel(const void *f, nvship_dim3 grid,
					 nvship_dim3 block, void **args,
					 unsigned int shmem,
					 nvship_stream_t stream)
{
	BOOTSTRAP();
	BUMP(H_hipLaunchCooperativeKernel);
	CHECK_REAL(hipLaunchCooperativeKernel);
	GATED_LAUNCH2(H_hipLaunchCooperativeKernel, real.hipLaunchCooperativeKernel(f, grid, block, args,
						     shmem, stream));
}

nvshipError_t hipModuleLaunchKernel(nvship_function_t f, unsigned int gx,
				    unsigned int gy, unsigned int gz,
				    unsigned int bx, unsigned int by,
				    unsigned int bz, unsigned int shmem,
				    nvship_stream_t stream, void **params,
				    void **extra)
{
	BOOTSTRAP();
	BUMP(H_hipModuleLaunchKernel);
	CHECK_REAL(hipModuleLaunchKernel);
	GATED_LAUNCH2(H_hipModuleLaunchKernel, real.hipModuleLaunchKernel(f, gx, gy, gz, bx, by, bz,
						shmem, stream, params,
						extra));
}

nvshipError_t hipExtModuleLaunchKernel(nvship_function_t f, uint32_t gwx,
				       uint32_t gwy, uint32_t gwz,
				       uint32_t lwx, uint32_t lwy,
				       uint32_t lwz, size_t shmem,
				       nvship_stream_t stream,
				       void **params, void **extra,
				       nvship_event_t ev0,
				       nvship_event_t ev1, uint32_t flags)
{
	BOOTSTRAP();
	BUMP(H_hipExtModuleLaunchKernel);
	CHECK_REAL(hipExtModuleLaunchKernel);
	GATED_LAUNCH2(H_hipExtModuleLaunchKernel, real.hipExtModuleLaunchKernel(f, gwx, gwy, gwz, lwx,
						   lwy, lwz, shmem, stream,
						   params, extra, ev0, ev1,
						   flags));
}

nvshipError_t hipGraphLaunch(nvship_graphexec_t g, nvship_stream_t stream)
{
	BOOTSTRAP();
	BUMP(H_hipGraphLaunch);
	CHECK_REAL(hipGraphLaunch);
	GATED_LAUNCH2(H_hipGraphLaunch, real.hipGraphLaunch(g, stream));
}

nvshipError_t hipMemcpy(void *dst, const void *src, size_t n,
			nvship_memcpy_kind kind)
{
	BOOTSTRAP();
	BUMP(H_hipMemcpy);
	CHECK_REAL(hipMemcpy);
	GATED_COPY(H_hipMemcpy, dst, n, NULL,
		   real.hipMemcpy(dst, src, n, kind));
}

nvshipError_t hipMemcpyAsync(void *dst, const void *src, size_t n,
			     nvship_memcpy_kind kind, nvship_stream_t s)
{
	BOOTSTRAP();
	BUMP(H_hipMemcpyAsync);
	CHECK_REAL(hipMemcpyAsync);
	GATED_COPY(H_hipMemcpyAsync, dst, n, s,
		   real.hipMemcpyAsync(dst, src, n, kind, s));
}

nvshipError_t hipMemcpyWithStream(void *dst, const void *src, size_t n,
				  nvship_memcpy_kind kind,
				  nvship_stream_t s)
{
	BOOTSTRAP();
	BUMP(H_hipMemcpyWithStream);
	CHECK_REAL(hipMemcpyWithStream);
	GATED_COPY(H_hipMemcpyWithStream, dst, n, s,
		   real.hipMemcpyWithStream(dst, src, n, kind, s));
}

nvshipError_t hipMemcpyHtoD(nvship_deviceptr_t dst, const void *src,
			    size_t n)
{
	BOOTSTRAP();
	BUMP(H_hipMemcpyHtoD);
	CHECK_REAL(hipMemcpyHtoD);
	GATED_COPY(H_hipMemcpyHtoD, dst, n, NULL,
		   real.hipMemcpyHtoD(dst, src, n));
}

nvshipError_t hipMemcpyDtoH(void *dst, nvship_deviceptr_t src, size_t n)
{
	BOOTSTRAP();
	BUMP(H_hipMemcpyDtoH);
	CHECK_REAL(hipMemcpyDtoH);
	GATED2(H_hipMemcpyDtoH, real.hipMemcpyDtoH(dst, src, n));
}

nvshipError_t hipMemcpyDtoD(nvship_deviceptr_t dst, nvship_deviceptr_t src,
			    size_t n)
{
	BOOTSTRAP();
	BUMP(H_hipMemcpyDtoD);
	CHECK_REAL(hipMemcpyDtoD);
	GATED2(H_hipMemcpyDtoD, real.hipMemcpyDtoD(dst, src, n));
}

nvshipError_t hipMemcpyHtoDAsync(nvship_deviceptr_t dst, const void *src,
				 size_t n, nvship_stream_t s)
{
	BOOTSTRAP();
	BUMP(H_hipMemcpyHtoDAsync);
	CHECK_REAL(hipMemcpyHtoDAsync);
	GATED_COPY(H_hipMemcpyHtoDAsync, dst, n, s,
		   real.hipMemcpyHtoDAsync(dst, src, n, s));
}

nvshipError_t hipMemcpyDtoHAsync(void *dst, nvship_deviceptr_t src,
				 size_t n, nvship_stream_t s)
{
	BOOTSTRAP();
	BUMP(H_hipMemcpyDtoHAsync);
	CHECK_REAL(hipMemcpyDtoHAsync);
	GATED2(H_hipMemcpyDtoHAsync, real.hipMemcpyDtoHAsync(dst, src, n, s));
}

nvshipError_t hipMemcpyDtoDAsync(nvship_deviceptr_t dst,
				 nvship_deviceptr_t src, size_t n,
				 nvship_stream_t s)
{
	BOOTSTRAP();
	BUMP(H_hipMemcpyDtoDAsync);
	CHECK_REAL(hipMemcpyDtoDAsync);
	GATED2(H_hipMemcpyDtoDAsync, real.hipMemcpyDtoDAsync(dst, src, n, s));
}

nvshipError_t hipMemset(void *dst, int value, size_t n)
{
	BOOTSTRAP();
	BUMP(H_hipMemset);
	CHECK_REAL(hipMemset);
	GATED_COPY(H_hipMemset, dst, n, NULL,
		   real.hipMemset(dst, value, n));
}

nvshipError_t hipMemsetAsync(void *dst, int value, size_t n,
			     nvship_stream_t s)
{
	BOOTSTRAP();
	BUMP(H_hipMemsetAsync);
	CHECK_REAL(hipMemsetAsync);
	GATED_COPY(H_hipMemsetAsync, dst, n, s,
		   real.hipMemsetAsync(dst, value, n, s));
}

nvshipError_t hipMemsetD32Async(nvship_deviceptr_t dst, int value,
				size_t count, nvship_stream_t s)
{
	BOOTSTRAP();
	BUMP(H_hipMemsetD32Async);
	CHECK_REAL(hipMemsetD32Async);
	GATED_COPY(H_hipMemsetD32Async, dst, count * 4, s,
		   real.hipMemsetD32Async(dst, value, count, s));
}

nvshipError_t hipMemcpy2D(void *dst, size_t dpitch, const void *src,
			  size_t spitch, size_t width, size_t height,
			  nvship_memcpy_kind kind)
{
	BOOTSTRAP();
	BUMP(H_hipMemcpy2D);
	CHECK_REAL(hipMemcpy2D);
	GATED2(H_hipMemcpy2D, real.hipMemcpy2D(dst, dpitch, src, spitch, width, height,
			       kind));
}

nvshipError_t hipMemcpy2DAsync(void *dst, size_t dpitch, const void *src,
			       size_t spitch, size_t width, size_t height,
			       nvship_memcpy_kind kind, nvship_stream_t s)
{
	BOOTSTRAP();
	BUMP(H_hipMemcpy2DAsync);
	CHECK_REAL(hipMemcpy2DAsync);
	GATED2(H_hipMemcpy2DAsync, real.hipMemcpy2DAsync(dst, dpitch, src, spitch, width,
				    height, kind, s));
}

nvshipError_t hipMemcpyToSymbol(const void *symbol, const void *src,
				size_t n, size_t offset,
				nvship_memcpy_kind kind)
{
	BOOTSTRAP();
	BUMP(H_hipMemcpyToSymbol);
	CHECK_REAL(hipMemcpyToSymbol);
	GATED2(H_hipMemcpyToSymbol, real.hipMemcpyToSymbol(symbol, src, n, offset, kind));
}

nvshipError_t hipMemcpyFromSymbol(void *dst, const void *symbol, size_t n,
				  size_t offset, nvship_memcpy_kind kind)
{
	BOOTSTRAP();
	BUMP(H_hipMemcpyFromSymbol);
	CHECK_REAL(hipMemcpyFromSymbol);
	GATED2(H_hipMemcpyFromSymbol, real.hipMemcpyFromSymbol(dst, symbol, n, offset, kind));
}

nvshipError_t hipMemcpyPeerAsync(void *dst, int dst_dev, const void *src,
				 int src_dev, size_t n, nvship_stream_t s)
{
	BOOTSTRAP();
	BUMP(H_hipMemcpyPeerAsync);
	CHECK_REAL(hipMemcpyPeerAsync);
	GATED2(H_hipMemcpyPeerAsync, real.hipMemcpyPeerAsync(dst, dst_dev, src, src_dev, n, s));
}

nvshipError_t hipModuleLaunchCooperativeKernel(
	nvship_function_t f, unsigned int gx, unsigned int gy,
	unsigned int gz, unsigned int bx, unsigned int by, unsigned int bz,
	unsigned int shmem, nvship_stream_t stream, void **params)
{
	BOOTSTRAP();
	BUMP(H_hipModuleLaunchCooperativeKernel);
	CHECK_REAL(hipModuleLaunchCooperativeKernel);
	GATED_LAUNCH2(H_hipModuleLaunchCooperativeKernel, real.hipModuleLaunchCooperativeKernel(
		f, gx, gy, gz, bx, by, bz, shmem, stream, params));
}

/*
 * Virtual-memory-management path (PyTorch
 * PYTORCH_HIP_ALLOC_CONF=expandable_segments:True allocates via
 * hipMemCreate/hipMemMap and would otherwise bypass every invariant;
 * the reference never faced it — CUDA 11.1-era PyTorch, SURVEY.md §7
 * step 4).  Physical VMM handles cannot be converted to managed
 * memory, so these allocations stay real VRAM: they are COUNTED
 * against the cap (handle-keyed) and loudly flagged as
 * non-oversubscribable.  client_env() additionally strips
 * expandable_segments from the allocator config so PyTorch under the
 * device plugin never takes this path by default.
 */
struct nvs_vmm_alloc {
	nvship_memhandle_t handle;
	size_t size;
	struct nvs_vmm_alloc *next;
};
static struct nvs_vmm_alloc *vmm_list;
static pthread_mutex_t vmm_mutex = PTHREAD_MUTEX_INITIALIZER;

nvshipError_t hipMemCreate(nvship_memhandle_t *handle, size_t size,
			   const void *prop, unsigned long long flags)
{
	static int warned;
	nvshipError_t r;

	BOOTSTRAP();
	BUMP(H_hipMemCreate);
	CHECK_REAL(hipMemCreate);
	if (!disable_um && !warned) {
		warned = 1;
		log_warn("hipMemCreate (VMM / expandable segments) in use: "
			 "these allocations are capped but stay in real "
			 "VRAM and cannot be oversubscribed or migrated; "
			 "prefer the default PyTorch allocator under "
			 "nvshare");
	}
	if (!disable_um && !oversub_allowed && size > 0 &&
	    reserve_cap(size) != 0)
		return NVSHIP_ERROR_OOM;
	r = real.hipMemCreate(handle, size, prop, flags);
	if (r == NVSHIP_SUCCESS && !disable_um && size > 0) {
		struct nvs_vmm_alloc *a = malloc(sizeof(*a));

		if (a != NULL) {
			a->handle = *handle;
			a->size = size;
			pthread_mutex_lock(&vmm_mutex);
			a->next = vmm_list;
			vmm_list = a;
			pthread_mutex_unlock(&vmm_mutex);
			if (oversub_allowed) {
				pthread_mutex_lock(&alloc_mutex);
				sum_allocated += size;
				pthread_mutex_unlock(&alloc_mutex);
			}
			log_debug("hook: +vmm %llu %zu MiB (sum %zu MiB)",
				  (unsigned long long)*handle,
				  size / NVS_MIB, sum_allocated / NVS_MIB);
		}
	} else if (r != NVSHIP_SUCCESS && !disable_um && !oversub_allowed &&
		   size > 0) {
		unreserve_cap(size);
	}
	return r;
}

nvshipError_t hipMemRelease(nvship_memhandle_t handle)
{
	struct nvs_vmm_alloc **pp, *a;
	size_t size = 0;

	BOOTSTRAP();
	BUMP(H_hipMemRelease);
	CHECK_REAL(hipMemRelease);
	pthread_mutex_lock(&vmm_mutex);
	for (pp = &vmm_list; *pp != NULL; pp = &(*pp)->next) {
		if ((*pp)->handle == handle) {
			a = *pp;
			*pp = a->next;
			size = a->size;
			free(a);
			break;
		}
	}
	pthread_mutex_unlock(&vmm_mutex);
	if (size != 0) {
		pthread_mutex_lock(&alloc_mutex);
		sum_allocated -= size;
		pthread_mutex_unlock(&alloc_mutex);
		log_debug("hook: -vmm %llu %zu MiB (sum %zu MiB)",
			  (unsigned long long)handle, size / NVS_MIB,
			  sum_allocated / NVS_MIB);
	}
	return real.hipMemRelease(handle);
}

nvshipError_t hipMemMap(void *ptr, size_t size, size_t offset,
			nvship_memhandle_t handle, unsigned long long flags)
{
	BOOTSTRAP();
	BUMP(H_hipMemMap);
	CHECK_REAL(hipMemMap);
	return real.hipMemMap(ptr, size, offset, handle, flags);
}

/* Direct managed allocations by the app: tracked and capped too. */
nvshipError_t hipMallocManaged(void **ptr, size_t size, unsigned int flags)
{
	nvshipError_t r;

	BOOTSTRAP();
	BUMP(H_hipMallocManaged);
	CHECK_REAL(hipMallocManaged);
	if (ptr == NULL)
		return NVSHIP_ERROR_INVALID_VALUE;
	if (!disable_um && !oversub_allowed && size > 0 &&
	    reserve_cap(size) != 0)
		return NVSHIP_ERROR_OOM;
	r = real.hipMallocManaged(ptr, size, flags);
	if (r == NVSHIP_SUCCESS && size > 0 && !disable_um)
		track_alloc(*ptr, size, 1, !oversub_allowed);
	else if (r != NVSHIP_SUCCESS && !disable_um && !oversub_allowed &&
		 size > 0)
		unreserve_cap(size);
	return r;
}

/* ------------------------------------------------------------------ */
/* Entry-point rerouting: hipGetProcAddress + versioned dlsym           */
/* ------------------------------------------------------------------ */

nvshipError_t hipGetProcAddress(const char *symbol, void **pfn,
				int hip_version, uint64_t flags,
				void *symbol_status);

struct hook_entry {
	const char *name;
	void *fn;
};

static const struct hook_entry hook_table[] = {
	{ "hipMalloc", (void *)hipMalloc },
	{ "hipExtMallocWithFlags", (void *)hipExtMallocWithFlags },
	{ "hipMallocAsync", (void *)hipMallocAsync },
	{ "hipMallocFromPoolAsync", (void *)hipMallocFromPoolAsync },
	{ "hipFree", (void *)hipFree },
	{ "hipFreeAsync", (void *)hipFreeAsync },
	{ "hipMemGetInfo", (void *)hipMemGetInfo },
	{ "hipSetDevice", (void *)hipSetDevice },
	{ "hipLaunchKernel", (void *)hipLaunchKernel },
	{ "hipExtLaunchKernel", (void *)hipExtLaunchKernel },
	{ "hipLaunchCooperativeKernel",
	  (void *)hipLaunchCooperativeKernel },
	{ "hipModuleLaunchKernel", (void *)hipModuleLaunchKernel },
	{ "hipExtModuleLaunchKernel", (void *)hipExtModuleLaunchKernel },
	{ "hipGraphLaunch", (void *)hipGraphLaunch },
	{ "hipMemcpy", (void *)hipMemcpy },
	{ "hipMemcpyAsync", (void *)hipMemcpyAsync },
	{ "hipMemcpyWithStream", (void *)hipMemcpyWithStream },
	{ "hipMemcpyHtoD", (void *)hipMemcpyHtoD },
	{ "hipMemcpyDtoH", (void *)hipMemcpyDtoH },
	{ "hipMemcpyDtoD", (void *)hipMemcpyDtoD },
	{ "hipMemcpyHtoDAsync", (void *)hipMemcpyHtoDAsync },
	{ "hipMemcpyDtoHAsync", (void *)hipMemcpyDtoHAsync },
	{ "hipMemcpyDtoDAsync", (void *)hipMemcpyDtoDAsync },
	{ "hipMemset", (void *)hipMemset },
	{ "hipMemsetAsync", (void *)hipMemsetAsync },
	{ "hipMemsetD32Async", (void *)hipMemsetD32Async },
	{ "hipMemcpy2D", (void *)hipMemcpy2D },
	{ "hipMemcpy2DAsync", (void *)hipMemcpy2DAsync },
	{ "hipMemcpyToSymbol", (void *)hipMemcpyToSymbol },
	{ "hipMemcpyFromSymbol", (void *)hipMemcpyFromSymbol },
	{ "hipMemcpyPeerAsync", (void *)hipMemcpyPeerAsync },
	{ "hipModuleLaunchCooperativeKernel",
	  (void *)hipModuleLaunchCooperativeKernel },
	{ "hipMallocManaged", (void *)hipMallocManaged },
	{ "hipMemCreate", (void *)hipMemCreate },
	{ "hipMemRelease", (void *)hipMemRelease },
	{ "hipMemMap", (void *)hipMemMap },
	/* dlsym-resolved hipGetProcAddress must return OUR wrapper,
	 * else every entry point fetched through it silently bypasses
	 * the cap and the gate (the reference hooked its
	 * cuGetProcAddress equivalent in the dlsym path too). */
	{ "hipGetProcAddress", (void *)hipGetProcAddress },
	{ NULL, NULL },
};

static void *lookup_hook(const char *name)
{
	const struct hook_entry *e;

	if (name == NULL || name[0] != 'h')
		return NULL;
	for (e = hook_table; e->name != NULL; e++)
		if (strcmp(e->name, name) == 0)
			return e->fn;
	return NULL;
}

nvshipError_t hipGetProcAddress(const char *symbol, void **pfn,
				int hip_version, uint64_t flags,
				void *symbol_status)
{
	nvshipError_t r;
	void *ours;

	BOOTSTRAP();
	BUMP(H_hipGetProcAddress);
	CHECK_REAL(hipGetProcAddress);
	r = real.hipGetProcAddress(symbol, pfn, hip_version, flags,
				   symbol_status);
	ours = lookup_hook(symbol);
	if (r == NVSHIP_SUCCESS && ours != NULL && pfn != NULL &&
	    *pfn != NULL)
		*pfn = ours;
	return r;
}

/*
 * Versioned dlsym interposition for dlopen-style loaders (mirrors the
 * reference's tactic, hook.c:346-415,974-975): glibc < 2.34 binds
 * dlsym@GLIBC_2.2.5 from libdl, >= 2.34 binds dlsym@GLIBC_2.34 from
 * libc; we export both and forward via dlvsym.
 */
static void *nvs_dlsym_common(void *handle, const char *name)
{
	void *ours = lookup_hook(name);

	if (ours != NULL) {
		/* Only reroute when the real library would satisfy it. */
		void *theirs = real_dlsym(handle, name);

		return theirs != NULL ? ours : NULL;
	}
	return real_dlsym(handle, name);
}

void *nvs_dlsym_234(void *handle, const char *name)
{
	return nvs_dlsym_common(handle, name);
}

void *nvs_dlsym_225(void *handle, const char *name)
{
	return nvs_dlsym_common(handle, name);
}

/* TSan builds skip the dlsym@GLIBC export: ThreadSanitizer's own
 * dlsym interceptor collides with it before its runtime initializes
 * (build-tsan is for race checking the threading logic, not the
 * loader tricks). */
#ifndef NVSHARE_NO_DLSYM_EXPORT
__asm__(".symver nvs_dlsym_234, dlsym@@GLIBC_2.34");
__asm__(".symver nvs_dlsym_225, dlsym@GLIBC_2.2.5");
#endif
