/*
 * nvshare-amd client runtime: the in-app scheduling agent.
 *
 * Parity reference: /root/reference/src/client.c (client thread,
 * early-release thread, continue_with_lock gate, idle detection).
 * nvshare-amd differences:
 *   - submission read/write lock: app threads hold a read lock across
 *     the real HIP call, the drain path takes the write lock, so a
 *     DROP_LOCK drain cannot race in-flight submissions (the reference
 *     tolerated this race);
 *   - connect timeout with a clear error instead of hanging forever at
 *     the first GPU call (NVSHARE_CONNECT_TIMEOUT_S, default 30), plus
 *     NVSHARE_STANDALONE=1 to free-run without a scheduler (used to
 *     measure managed-memory overhead in isolation);
 *   - scheduler-restart tolerance: on socket death the client
 *     reconnects and re-registers for NVSHARE_RECONNECT_S (default 60)
 *     seconds before giving up (the reference killed the app);
 *   - idle detection via ROCm SMI busy% (librocm_smi64) with the
 *     reference's timed-synchronize fallback;
 *   - optional hipMemPrefetchAsync of tracked allocations on LOCK_OK
 *     (NVSHARE_PREFETCH=1): gfx950 HMM fault-retry is slower than bulk
 *     migration, so prefetching the working set back after a handoff
 *     cuts the refault storm.
 */
#define _GNU_SOURCE
#include <dlfcn.h>
#include <fcntl.h>
#include <inttypes.h>
#include <pthread.h>
#include <semaphore.h>
#include <signal.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <unistd.h>

#include "common.h"
#include "proto.h"
#include "hip_defs.h"
#include "client.h"

/* Provided by hook.c. */
struct nvs_real_hip real;

int nvs_app_device = 0;

/* ---- state (guarded by g_mutex unless noted) ---- */
static pthread_mutex_t g_mutex = PTHREAD_MUTEX_INITIALIZER;
static pthread_cond_t own_lock_cv = PTHREAD_COND_INITIALIZER;
static int scheduler_on = 1;
static int own_lock = 0;
static int need_lock = 0;
static int did_work = 0;
static int standalone = 0;
static uint64_t client_id = 0;
/* Scheduler's global memory-pressure verdict, from LOCK_OK "p=" data:
 * -1 unknown, 0 fits, 1 oversubscribed.  Drives migration assist when
 * NVSHARE_PREFETCH/NVSHARE_EVICT are not explicitly set. */
static int sched_pressure = -1;

/* Sharing metrics (atomic; dumped at exit when NVSHARE_DEBUG):
 * - wait_ns: app-thread time blocked at the gate waiting for a grant
 * - held_ns: lock tenure (grant .. release)
 * - grants/preempts/early_releases: lifecycle counters */
static int64_t metric_wait_ns;
static int64_t metric_held_ns;
static int64_t metric_grants;
static int64_t metric_preempts;
static int64_t metric_early_releases;
static int64_t grant_t0_ns; /* guarded by g_mutex */

#define METRIC_ADD(var, v) __atomic_fetch_add(&(var), (v), 					      __ATOMIC_RELAXED)

/* Migration-assist policy: an explicit env setting wins; else the
 * scheduler's pressure verdict; else the AUTO_MIGRATE default (which
 * self-gates on real free memory in hook.c). */
static int migrate_enabled(const char *env)
{
	if (getenv(env) != NULL)
		return nvs_env_bool(env, 1);
	if (__atomic_load_n(&sched_pressure, __ATOMIC_RELAXED) >= 0)
		return __atomic_load_n(&sched_pressure, __ATOMIC_RELAXED);
	return nvs_env_bool("NVSHARE_AUTO_MIGRATE", 1);
}

/* Submission lock: readers = app threads inside a real HIP call,
 * writer = the drain path (DROP_LOCK / early release). */
static pthread_rwlock_t submit_rwlock = PTHREAD_RWLOCK_INITIALIZER;

/* Socket (writes guarded by sock_mutex; reads only from client thread).
 * Never take g_mutex while holding sock_mutex. */
static pthread_mutex_t sock_mutex = PTHREAD_MUTEX_INITIALIZER;
static int sock_fd = -1;

static int shutting_down; /* atomic accesses only */

static inline int is_shutting_down(void)
{
	return __atomic_load_n(&shutting_down, __ATOMIC_RELAXED);
}

static sem_t init_done_sem;
static char pod_name[NVS_POD_NAME_LEN];
static char pod_namespace[NVS_POD_NS_LEN];
static int physical_gpu;  /* node-level GPU index this client shares */

/* The node-level index of the GPU this process uses: NVSHARE_GPU, or
 * the first entry of ROCR/HIP_VISIBLE_DEVICES (containers see exactly
 * one GPU), else 0. */
static int detect_physical_gpu(void)
{
	const char *v = getenv("NVSHARE_GPU");

	if (v == NULL)
		v = getenv("ROCR_VISIBLE_DEVICES");
	if (v == NULL)
		v = getenv("HIP_VISIBLE_DEVICES");
	if (v != NULL && v[0] >= '0' && v[0] <= '9')
		return atoi(v);
	return 0;
}

/* idle detection */
static fn_rsmi_init p_rsmi_init;
static fn_rsmi_dev_busy_percent_get p_rsmi_busy;
static int rsmi_ready = 0;
static uint32_t rsmi_dev_index = 0;
static long release_interval_ms = 5000;
static long idle_sync_threshold_ms = 100;

static int send_msg_type(uint8_t type)
{
	struct nvs_msg m;
	int ret;

	nvs_msg_init(&m, type, client_id, NULL);
	nvs_strlcpy(m.pod_name, pod_name, sizeof(m.pod_name));
	nvs_strlcpy(m.pod_namespace, pod_namespace, sizeof(m.pod_namespace));
	pthread_mutex_lock(&sock_mutex);
	ret = (sock_fd >= 0) ? nvs_send_msg(sock_fd, &m) : -1;
	pthread_mutex_unlock(&sock_mutex);
	if (ret != 0)
		log_debug("client: send %s failed", nvs_msg_type_str(type));
	return ret;
}

void nvs_submit_begin(void)
{
	for (;;) {
		pthread_rwlock_rdlock(&submit_rwlock);
		pthread_mutex_lock(&g_mutex);
		if (!scheduler_on || own_lock) {
			did_work = 1;
			pthread_mutex_unlock(&g_mutex);
			/* Deferred managed-page population happens here,
			 * under the lock, before the real submission. */
			nvs_populate_pending();
			return; /* read lock held */
		}
		pthread_rwlock_unlock(&submit_rwlock);
		{
			int64_t w0 = nvs_now_ns();

			for (;;) {
				struct timespec abs;

				if (!scheduler_on || own_lock)
					break;
				if (!need_lock) {
					need_lock = 1;
					/* sock_mutex nests under g_mutex
					 * (never the other way around). */
					if (send_msg_type(NVS_REQ_LOCK) != 0)
						need_lock = 0; /* retry */
				}
				/* Re-send REQ_LOCK if nothing arrives for
				 * 10 s: idempotent at the scheduler, and
				 * insurance against lost-wakeup bugs. */
				clock_gettime(CLOCK_REALTIME, &abs);
				abs.tv_sec += 10;
				if (pthread_cond_timedwait(&own_lock_cv,
							   &g_mutex,
							   &abs) != 0 &&
				    scheduler_on && !own_lock)
					need_lock = 0;
			}
			METRIC_ADD(metric_wait_ns, nvs_now_ns() - w0);
		}
		pthread_mutex_unlock(&g_mutex);
	}
}

void nvs_submit_end(void)
{
	pthread_rwlock_unlock(&submit_rwlock);
}

/* Racy read on purpose: used only to skip window bookkeeping when no
 * scheduler is gating this process. */
int nvs_scheduler_gating(void)
{
	return scheduler_on && !standalone;
}

int nvs_can_submit_now(void)
{
	return standalone || !scheduler_on || own_lock;
}

/* Drain all outstanding GPU work (caller must NOT hold g_mutex). */
static void drain_gpu(void)
{
	pthread_rwlock_wrlock(&submit_rwlock);
	if (!is_shutting_down() && real.hipSetDevice != NULL)
		real.hipSetDevice(nvs_app_device);
	if (!is_shutting_down() && real.hipDeviceSynchronize != NULL)
		real.hipDeviceSynchronize();
	pthread_rwlock_unlock(&submit_rwlock);
}

/* exit() has begun: the HIP runtime / ROCm SMI may already be torn
 * down, so the injected threads must stop touching them.  Registered
 * with atexit (runs before library destructors). */
static void on_process_exit(void)
{
	__atomic_store_n(&shutting_down, 1, __ATOMIC_RELAXED);
	if (nvs_debug_enabled && !standalone)
		log_debug("client sharing metrics: grants=%lld "
			  "preempts=%lld early_releases=%lld "
			  "held=%.1fs waited=%.1fs",
			  (long long)metric_grants,
			  (long long)metric_preempts,
			  (long long)metric_early_releases,
			  metric_held_ns / 1e9, metric_wait_ns / 1e9);
}

static void read_pod_identity(void)
{
	const char *v;

	v = getenv("NVSHARE_POD_NAME");
	if (v == NULL)
		v = getenv("HOSTNAME");
	nvs_strlcpy(pod_name, v != NULL ? v : "unknown", sizeof(pod_name));

	pod_namespace[0] = '\0';
	if (getenv("KUBERNETES_SERVICE_HOST") != NULL) {
		int fd = open("/var/run/secrets/kubernetes.io/"
			      "serviceaccount/namespace",
			      O_RDONLY | O_CLOEXEC);
		if (fd >= 0) {
			ssize_t n = read(fd, pod_namespace,
					 sizeof(pod_namespace) - 1);
			if (n > 0)
				pod_namespace[n] = '\0';
			close(fd);
		}
	}
	v = getenv("NVSHARE_POD_NAMESPACE");
	if (v != NULL)
		nvs_strlcpy(pod_namespace, v, sizeof(pod_namespace));
}

/* Connect + REGISTER + initial status. Returns fd, or -1. */
static int connect_and_register(void)
{
	char path[NVS_SOCK_PATH_MAX];
	struct nvs_msg m;
	int fd;

	if (nvs_scheduler_path(path) != 0)
		return -1;
	fd = nvs_connect_path(path);
	if (fd < 0)
		return -1;
	{
		char gpu_tag[NVS_MSG_DATA_LEN];

		snprintf(gpu_tag, sizeof(gpu_tag), "gpu%d", physical_gpu);
		nvs_msg_init(&m, NVS_REGISTER, 0, gpu_tag);
	}
	nvs_strlcpy(m.pod_name, pod_name, sizeof(m.pod_name));
	nvs_strlcpy(m.pod_namespace, pod_namespace, sizeof(m.pod_namespace));
	if (nvs_send_msg(fd, &m) != 0) {
		close(fd);
		return -1;
	}
	if (nvs_recv_msg(fd, &m) != 0 ||
	    (m.type != NVS_SCHED_ON && m.type != NVS_SCHED_OFF)) {
		close(fd);
		return -1;
	}
	pthread_mutex_lock(&g_mutex);
	scheduler_on = (m.type == NVS_SCHED_ON);
	own_lock = 0;
	need_lock = 0;
	{
		char buf[NVS_MSG_DATA_LEN];
		memcpy(buf, m.data, NVS_MSG_DATA_LEN);
		buf[NVS_MSG_DATA_LEN - 1] = '\0';
		client_id = strtoull(buf, NULL, 16);
	}
	pthread_cond_broadcast(&own_lock_cv);
	pthread_mutex_unlock(&g_mutex);
	log_debug("client: registered id=%016" PRIx64 " (scheduling %s)",
		  client_id, scheduler_on ? "on" : "off");
	return fd;
}

static void handle_lock_ok(const struct nvs_msg *m)
{
	/* The scheduler includes its whole-node pressure verdict with
	 * the grant ("p=0"/"p=1"); reference-era schedulers send no
	 * data, leaving the verdict unknown. */
	if (m->data[0] == 'p' && m->data[1] == '=')
		__atomic_store_n(&sched_pressure, m->data[2] == '1' ? 1 : 0,
				 __ATOMIC_RELAXED);
	/* Restore the working set BEFORE opening the gate: demand
	 * faults racing the bulk migration degrade both.  Under
	 * pressure (or with NVSHARE_PREFETCH=1 / the AUTO_MIGRATE
	 * fallback) restore on grant — when the set is already resident
	 * the prefetch is a no-op costing microseconds; when it was
	 * evicted this is 25x faster than demand refaulting
	 * (profiles/restorebench.json). */
	if (migrate_enabled("NVSHARE_PREFETCH"))
		nvs_prefetch_allocs();
	pthread_mutex_lock(&g_mutex);
	own_lock = 1;
	need_lock = 0;
	grant_t0_ns = nvs_now_ns();
	METRIC_ADD(metric_grants, 1);
	pthread_cond_broadcast(&own_lock_cv);
	pthread_mutex_unlock(&g_mutex);
}

static void handle_drop_lock(void)
{
	int had;

	pthread_mutex_lock(&g_mutex);
	had = own_lock;
	own_lock = 0;
	if (had) {
		METRIC_ADD(metric_held_ns, nvs_now_ns() - grant_t0_ns);
		METRIC_ADD(metric_preempts, 1);
	}
	pthread_mutex_unlock(&g_mutex);
	if (!had)
		return; /* already released voluntarily */
	drain_gpu();
	/* NVSHARE_EVICT_ASYNC=1 (experiment, default off): release the
	 * lock BEFORE evicting so the next holder's restore overlaps
	 * our eviction — cuts the handoff critical path from
	 * evict+restore to max(evict, restore) IF the two bulk
	 * migrations don't contend (unvalidated on hardware; the
	 * round-1 livelock was demand faults racing bulk migration,
	 * a different pair). */
	if (nvs_env_bool("NVSHARE_EVICT_ASYNC", 0)) {
		send_msg_type(NVS_LOCK_RELEASED);
		if (migrate_enabled("NVSHARE_EVICT"))
			nvs_evict_allocs();
		log_debug("client: lock released (async evict)");
		return;
	}
	/* Eviction additionally self-gates on real memory pressure (it
	 * is skipped when free HBM already fits the tracked set), so the
	 * automatic default is safe for fitting workloads. */
	if (migrate_enabled("NVSHARE_EVICT"))
		nvs_evict_allocs(); /* blocking; under pressure only */
	send_msg_type(NVS_LOCK_RELEASED);
	log_debug("client: lock released after DROP_LOCK");
}

static void handle_sched_status(int on)
{
	pthread_mutex_lock(&g_mutex);
	scheduler_on = on;
	own_lock = 0;
	need_lock = 0;
	pthread_cond_broadcast(&own_lock_cv);
	pthread_mutex_unlock(&g_mutex);
	log_debug("client: scheduling turned %s", on ? "on" : "off");
}

static void *client_thread(void *arg)
{
	sigset_t all;
	long reconnect_s;
	struct nvs_msg m;

	(void)arg;
	sigfillset(&all);
	pthread_sigmask(SIG_BLOCK, &all, NULL);

	reconnect_s = nvs_env_long("NVSHARE_RECONNECT_S", 60, 0, 86400);

	for (;;) {
		if (nvs_recv_msg(sock_fd, &m) != 0) {
			/* Scheduler died: try to reconnect. */
			double deadline = nvs_now_s() + (double)reconnect_s;
			int fd = -1;

			log_warn("client: lost scheduler connection, "
				 "reconnecting for up to %lds", reconnect_s);
			pthread_mutex_lock(&sock_mutex);
			if (sock_fd >= 0)
				close(sock_fd);
			sock_fd = -1;
			pthread_mutex_unlock(&sock_mutex);
			while (nvs_now_s() < deadline) {
				fd = connect_and_register();
				if (fd >= 0)
					break;
				usleep(500 * 1000);
			}
			if (fd < 0)
				log_fatal("client: cannot reach "
					  "nvshare-scheduler after %lds; "
					  "aborting", reconnect_s);
			pthread_mutex_lock(&sock_mutex);
			sock_fd = fd;
			pthread_mutex_unlock(&sock_mutex);
			/* Re-request the lock if app threads are waiting. */
			pthread_mutex_lock(&g_mutex);
			if (need_lock) {
				need_lock = 0;
				pthread_cond_broadcast(&own_lock_cv);
			}
			pthread_mutex_unlock(&g_mutex);
			continue;
		}
		switch (m.type) {
		case NVS_LOCK_OK:
			handle_lock_ok(&m);
			break;
		case NVS_DROP_LOCK:
			handle_drop_lock();
			break;
		case NVS_SCHED_ON:
			handle_sched_status(1);
			break;
		case NVS_SCHED_OFF:
			handle_sched_status(0);
			break;
		default:
			log_debug("client: ignoring %s",
				  nvs_msg_type_str(m.type));
			break;
		}
	}
	return NULL;
}

static void idle_detect_init(void)
{
	void *h;

	rsmi_dev_index = (uint32_t)physical_gpu;

	h = dlopen("librocm_smi64.so.7", RTLD_LAZY | RTLD_LOCAL);
	if (h == NULL)
		h = dlopen("librocm_smi64.so", RTLD_LAZY | RTLD_LOCAL);
	if (h == NULL) {
		log_debug("client: librocm_smi64 unavailable, using timed-"
			  "sync idle fallback");
		return;
	}
	p_rsmi_init = (fn_rsmi_init)dlsym(h, "rsmi_init");
	p_rsmi_busy = (fn_rsmi_dev_busy_percent_get)
		dlsym(h, "rsmi_dev_busy_percent_get");
	if (p_rsmi_init == NULL || p_rsmi_busy == NULL)
		return;
	if (p_rsmi_init(0) != 0) /* RSMI_STATUS_SUCCESS == 0 */
		return;
	rsmi_ready = 1;
	log_debug("client: ROCm SMI idle detection on device %u",
		  rsmi_dev_index);
}

/* 1 = GPU looks idle, 0 = busy/unknown. */
static int check_idle(void)
{
	if (is_shutting_down())
		return 0;
	/* rsmi busy%% is windowed and lags work completion by hundreds
	 * of ms — fine for the reference's 5 s cadence, but it IS the
	 * handoff latency when probing fast.  Sub-second cadences use
	 * the precise timed-sync probe instead (an idle device answers
	 * hipDeviceSynchronize in microseconds). */
	if (rsmi_ready && release_interval_ms >= 1000) {
		uint32_t busy = 100;

		if (p_rsmi_busy(rsmi_dev_index, &busy) == 0)
			return busy == 0;
		/* fall through to timed sync */
	}
	if (real.hipSetDevice != NULL && real.hipDeviceSynchronize != NULL) {
		int64_t t0 = nvs_now_ns(), dt_ms;

		real.hipSetDevice(nvs_app_device);
		real.hipDeviceSynchronize();
		dt_ms = (nvs_now_ns() - t0) / 1000000;
		return dt_ms < idle_sync_threshold_ms;
	}
	return 0;
}

static void *early_release_thread(void *arg)
{
	sigset_t all;

	(void)arg;
	sigfillset(&all);
	pthread_sigmask(SIG_BLOCK, &all, NULL);

	idle_detect_init();

	long last_mem_mib = -1;

	for (;;) {
		int worked, have_lock;
		long mem_mib;

		usleep((useconds_t)(release_interval_ms * 1000));
		if (is_shutting_down())
			continue; /* never touch HIP during teardown */
		mem_mib = nvs_sum_allocated_mib();
		if (mem_mib != last_mem_mib) {
			struct nvs_msg m;
			char buf[NVS_MSG_DATA_LEN];

			/* "mem,cap": capacity feeds the scheduler's
			 * pressure policy.  The 20-byte data field fits
			 * both for any real MiB value. */
			snprintf(buf, sizeof(buf), "%ld,%ld", mem_mib,
				 nvs_mem_total_mib());
			nvs_msg_init(&m, NVS_MEM_UPDATE, client_id, buf);
			pthread_mutex_lock(&sock_mutex);
			if (sock_fd >= 0 &&
			    nvs_send_msg(sock_fd, &m) == 0)
				last_mem_mib = mem_mib;
			pthread_mutex_unlock(&sock_mutex);
		}
		pthread_mutex_lock(&g_mutex);
		have_lock = own_lock && scheduler_on;
		worked = did_work;
		did_work = 0;
		pthread_mutex_unlock(&g_mutex);
		if (!have_lock || worked)
			continue;
		if (!check_idle())
			continue;
		/* Idle while holding the lock: hand the GPU back. */
		pthread_mutex_lock(&g_mutex);
		if (!own_lock || !scheduler_on) {
			pthread_mutex_unlock(&g_mutex);
			continue;
		}
		own_lock = 0;
		METRIC_ADD(metric_held_ns, nvs_now_ns() - grant_t0_ns);
		METRIC_ADD(metric_early_releases, 1);
		pthread_mutex_unlock(&g_mutex);
		drain_gpu();
		/* Same pressure-gated eviction as the DROP_LOCK path:
		 * leaving the idle set resident makes the next holder's
		 * restore fight fault-driven eviction. */
		if (migrate_enabled("NVSHARE_EVICT"))
			nvs_evict_allocs();
		send_msg_type(NVS_LOCK_RELEASED);
		log_debug("client: early-released idle lock");
	}
	return NULL;
}

/*
 * fork() safety: a forked child (e.g. a DataLoader worker) inherits
 * the preload state but not our threads, and shares the parent's
 * scheduler socket fd — writing to it would corrupt the parent's
 * protocol stream.  The child therefore detaches: fresh (unlocked)
 * synchronization objects and free-running standalone mode.  GPU work
 * from forked children is NOT arbitrated (same practical stance as the
 * reference, which would deadlock instead); CUDA-style spawn children
 * bootstrap their own client as usual.
 */
static void atfork_child(void)
{
	pthread_mutex_t fresh_m = PTHREAD_MUTEX_INITIALIZER;
	pthread_cond_t fresh_c = PTHREAD_COND_INITIALIZER;
	pthread_rwlock_t fresh_rw = PTHREAD_RWLOCK_INITIALIZER;

	memcpy(&g_mutex, &fresh_m, sizeof(fresh_m));
	memcpy(&sock_mutex, &fresh_m, sizeof(fresh_m));
	memcpy(&own_lock_cv, &fresh_c, sizeof(fresh_c));
	memcpy(&submit_rwlock, &fresh_rw, sizeof(fresh_rw));
	if (sock_fd >= 0)
		close(sock_fd);
	sock_fd = -1;
	standalone = 1;
	scheduler_on = 0;
	own_lock = 0;
	need_lock = 0;
	nvs_free_cache_forget();
}

void nvs_client_init(void)
{
	pthread_t tid;
	long timeout_s;
	double deadline;
	int fd = -1;

	pthread_atfork(NULL, NULL, atfork_child);
	atexit(on_process_exit);

	/* Millisecond-granularity idle probing: bursty/barrier-waiting
	 * clients hand the GPU over in ~2x this interval.  The coarser
	 * reference default was 5 s (client.c:51). */
	release_interval_ms = nvs_env_long("NVSHARE_RELEASE_INTERVAL_MS",
					   0, 0, 3600000);
	if (release_interval_ms == 0)
		release_interval_ms = nvs_env_long(
			"NVSHARE_RELEASE_INTERVAL_S", 5, 1, 3600) * 1000;
	idle_sync_threshold_ms = nvs_env_long("NVSHARE_IDLE_SYNC_MS", 100,
					      1, 60000);
	timeout_s = nvs_env_long("NVSHARE_CONNECT_TIMEOUT_S", 30, 1, 3600);
	standalone = nvs_env_bool("NVSHARE_STANDALONE", 0);

	sem_init(&init_done_sem, 0, 0);
	read_pod_identity();
	physical_gpu = detect_physical_gpu();

	if (standalone) {
		scheduler_on = 0;
		log_info("client: NVSHARE_STANDALONE=1, free-running "
			 "(managed memory only, no scheduling)");
		return;
	}

	deadline = nvs_now_s() + (double)timeout_s;
	while (nvs_now_s() < deadline) {
		fd = connect_and_register();
		if (fd >= 0)
			break;
		usleep(250 * 1000);
	}
	if (fd < 0) {
		char path[NVS_SOCK_PATH_MAX] = "?";

		nvs_scheduler_path(path);
		log_fatal("client: cannot reach nvshare-scheduler at %s "
			  "after %lds. Start nvshare-scheduler, or set "
			  "NVSHARE_STANDALONE=1 to run without scheduling.",
			  path, timeout_s);
	}
	sock_fd = fd;

	true_or_exit(pthread_create(&tid, NULL, client_thread, NULL) == 0);
	pthread_detach(tid);
	true_or_exit(pthread_create(&tid, NULL, early_release_thread, NULL)
		     == 0);
	pthread_detach(tid);
	log_info("client: connected to scheduler (id=%016" PRIx64 ")",
		 client_id);
}
