/*
 * nvshare-scheduler — per-node GPU arbiter for nvshare-amd (MI355X).
 *
 * Serializes GPU work across transparent-sharing clients with an FCFS
 * exclusive lock held for a time quantum (TQ), independently for every
 * GPU on the node.  Single-threaded epoll event loop plus one timer
 * thread.  Wire-compatible with the reference protocol (see proto.h);
 * behavioral parity reference: /root/reference/src/scheduler.c
 * (registry, FCFS queue, TQ timer, strict eviction, SCHED_ON/OFF
 * broadcast, SET_TQ).
 *
 * nvshare-amd additions over the reference:
 *   - multi-GPU: one daemon arbitrates all 8 GPUs of an MI355X node
 *     with independent per-GPU locks/queues; clients declare their GPU
 *     in REGISTER.data ("gpuN"; empty = GPU 0, so reference-shaped
 *     clients still work).  The reference hard-coded a single GPU
 *     (README.md:97);
 *   - env-configurable startup state: NVSHARE_TQ (seconds),
 *     NVSHARE_SCHED_OFF=1 (start with scheduling disabled)
 *     (reference left these as TODOs, scheduler.c:549-552);
 *   - solo-client fast path: the TQ timer does not preempt the lock
 *     holder when nobody else is queued on that GPU, so a lone client
 *     never pays periodic drain stalls; a newly arriving waiter
 *     preempts the holder as soon as it has had >= one full quantum;
 *   - STATUS_REQ/STATUS query for observability (nvsharectl -q);
 *   - partial-read tolerant framing (per-connection receive buffer);
 *   - SIGUSR1 dumps scheduler state to stderr.
 */
#define _GNU_SOURCE
#include <errno.h>
#include <fcntl.h>
#include <inttypes.h>
#include <pthread.h>
#include <signal.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/epoll.h>
#include <sys/socket.h>
#include <sys/stat.h>
#include <unistd.h>

#include "common.h"
#include "proto.h"

#define NVS_DEFAULT_TQ 30
#define MAX_EPOLL_EVENTS 64
#define MAX_GPUS 16

/* One Unix-socket connection. Becomes a "client" after REGISTER. */
struct conn {
	int fd;
	size_t got;               /* bytes of in-progress message */
	struct nvs_msg inmsg;
	int registered;
	uint64_t id;              /* client id (registered only) */
	int gpu;                  /* GPU this client shares */
	char pod_name[NVS_POD_NAME_LEN];
	char pod_namespace[NVS_POD_NS_LEN];
	int wants_lock;           /* present in its GPU's request queue */
	int relock_pending;       /* holder re-requested during release */
	int dead;                 /* unwritable; evict at a safe point */
	long mem_mib;             /* client-reported tracked allocations */
	long cap_mib;             /* client-reported device capacity */
	struct conn *next;        /* registry list */
	struct conn *qnext;       /* FCFS queue list */
};

/* Per-GPU arbitration state. */
struct gpu_state {
	struct conn *queue_head;  /* FCFS queue; head == holder */
	struct conn *queue_tail;
	int lock_held;
	struct conn *lock_holder;
	int drop_lock_sent;
	unsigned long round;
	int64_t quantum_start_ns;
	unsigned long grants, preemptions;
	int tq_override;          /* 0 = use the global tq */
};

static struct conn *clients;      /* all connections */
static struct gpu_state gpus[MAX_GPUS];

static pthread_mutex_t g_mutex = PTHREAD_MUTEX_INITIALIZER;
static pthread_cond_t timer_cv = PTHREAD_COND_INITIALIZER;

static int scheduler_on = 1;
static int tq_seconds = NVS_DEFAULT_TQ;
static unsigned long total_evictions;

static int epoll_fd = -1;
static char sock_path[NVS_SOCK_PATH_MAX];
static volatile sig_atomic_t dump_requested;

static int queue_len(const struct gpu_state *g)
{
	int n = 0;
	struct conn *c;

	for (c = g->queue_head; c != NULL; c = c->qnext)
		n++;
	return n;
}

static int queue_len_all(void)
{
	int n = 0, i;

	for (i = 0; i < MAX_GPUS; i++)
		n += queue_len(&gpus[i]);
	return n;
}

static int client_count(void)
{
	int n = 0;
	struct conn *c;

	for (c = clients; c != NULL; c = c->next)
		if (c->registered)
			n++;
	return n;
}

static void send_to(struct conn *c, uint8_t type, const char *data)
{
	struct nvs_msg m;

	if (c->dead)
		return; /* already unwritable; sweep will evict it */
	nvs_msg_init(&m, type, c->id, data);
	if (nvs_send_msg(c->fd, &m) != 0) {
		/* Framing toward this peer can no longer be trusted:
		 * mark it for eviction at the end of the event-loop
		 * pass (deleting here would invalidate pointers the
		 * callers still hold). */
		log_warn("send %s to client %016" PRIx64 " failed; "
			 "marking for eviction", nvs_msg_type_str(type),
			 c->id);
		c->dead = 1;
	} else
		log_debug("sent %s to client %016" PRIx64,
			  nvs_msg_type_str(type), c->id);
}

static void queue_push(struct gpu_state *g, struct conn *c)
{
	c->qnext = NULL;
	if (g->queue_tail != NULL)
		g->queue_tail->qnext = c;
	else
		g->queue_head = c;
	g->queue_tail = c;
	c->wants_lock = 1;
}

static void queue_remove(struct gpu_state *g, struct conn *c)
{
	struct conn **pp = &g->queue_head;
	struct conn *prev = NULL, *it;

	for (it = g->queue_head; it != NULL; prev = it, it = it->qnext) {
		if (it == c) {
			*pp = it->qnext;
			if (g->queue_tail == c)
				g->queue_tail = prev;
			break;
		}
		pp = &it->qnext;
	}
	c->qnext = NULL;
	c->wants_lock = 0;
}

/*
 * Global memory-pressure verdict for one GPU: 1 when the clients'
 * combined tracked working sets exceed the device capacity they
 * report, 0 when they fit, -1 when no client has reported a capacity
 * yet.  Broadcast with LOCK_OK ("p=0"/"p=1") so clients flip their
 * migration assist (evict-on-release / prefetch-on-grant) on or off
 * from the scheduler's whole-node view instead of per-process env
 * guesses (docs/roadmap.md #2).
 */
static int gpu_pressure(const struct gpu_state *g)
{
	const struct conn *c;
	long sum = 0, cap = 0;
	int gi = (int)(g - gpus);

	for (c = clients; c != NULL; c = c->next) {
		if (!c->registered || c->gpu != gi)
			continue;
		sum += c->mem_mib;
		if (c->cap_mib > cap)
			cap = c->cap_mib;
	}
	if (cap <= 0)
		return -1;
	return sum > cap;
}

/* Grant GPU g's lock to its queue head if possible. Holds g_mutex. */
static void try_schedule(struct gpu_state *g)
{
	int pressure;
	const char *pdata;

	if (!scheduler_on || g->lock_held || g->queue_head == NULL)
		return;
	g->lock_holder = g->queue_head;
	g->lock_held = 1;
	g->drop_lock_sent = 0;
	g->round++;
	g->grants++;
	g->quantum_start_ns = nvs_now_ns();
	pressure = gpu_pressure(g);
	pdata = pressure < 0 ? NULL : (pressure ? "p=1" : "p=0");
	send_to(g->lock_holder, NVS_LOCK_OK, pdata);
	log_debug("gpu%ld round %lu: lock -> %016" PRIx64 " (queue=%d)",
		  (long)(g - gpus), g->round, g->lock_holder->id,
		  queue_len(g));
	pthread_cond_broadcast(&timer_cv);
}

/* A GPU is "armed" when its holder can be preempted: lock held, at
 * least one waiter behind the holder, DROP_LOCK not yet sent. */
static int gpu_tq(const struct gpu_state *g)
{
	return g->tq_override > 0 ? g->tq_override : tq_seconds;
}

static int gpu_armed(const struct gpu_state *g)
{
	return g->lock_held && scheduler_on && !g->drop_lock_sent &&
	       g->queue_head != NULL && g->queue_head->qnext != NULL;
}

/*
 * TQ timer thread.  For every armed GPU, send DROP_LOCK once the
 * holder has had tq seconds.  A lone holder is never preempted (solo
 * fast path); a waiter's arrival wakes this thread, which preempts as
 * soon as the holder's quantum is exhausted.  After DROP_LOCK we wait
 * indefinitely for LOCK_RELEASED; only socket death evicts a stuck
 * client (strict eviction, as the reference: scheduler.c:352,645-663).
 */
static void *timer_thread(void *arg)
{
	(void)arg;
	pthread_mutex_lock(&g_mutex);
	for (;;) {
		int64_t now = nvs_now_ns();
		int64_t next_deadline = INT64_MAX;
		int i, any_armed = 0;

		for (i = 0; i < MAX_GPUS; i++) {
			struct gpu_state *g = &gpus[i];
			int64_t deadline;

			if (!gpu_armed(g))
				continue;
			any_armed = 1;
			deadline = g->quantum_start_ns +
				   (int64_t)gpu_tq(g) * 1000000000LL;
			if (deadline <= now) {
				g->drop_lock_sent = 1;
				g->preemptions++;
				log_debug("gpu%d round %lu: TQ expired, "
					  "DROP_LOCK -> %016" PRIx64, i,
					  g->round, g->lock_holder->id);
				send_to(g->lock_holder, NVS_DROP_LOCK,
					NULL);
				continue;
			}
			if (deadline < next_deadline)
				next_deadline = deadline;
		}

		if (!any_armed || next_deadline == INT64_MAX) {
			pthread_cond_wait(&timer_cv, &g_mutex);
			continue;
		}
		{
			struct timespec abs;
			int64_t wait_ns = next_deadline - nvs_now_ns();

			if (wait_ns <= 0)
				continue;
			clock_gettime(CLOCK_REALTIME, &abs);
			abs.tv_sec += wait_ns / 1000000000LL;
			abs.tv_nsec += wait_ns % 1000000000LL;
			if (abs.tv_nsec >= 1000000000L) {
				abs.tv_sec++;
				abs.tv_nsec -= 1000000000L;
			}
			pthread_cond_timedwait(&timer_cv, &g_mutex, &abs);
		}
	}
	return NULL;
}

static void bcast_status(void)
{
	struct conn *c;

	for (c = clients; c != NULL; c = c->next)
		if (c->registered)
			send_to(c, scheduler_on ? NVS_SCHED_ON :
						  NVS_SCHED_OFF, NULL);
}

static void delete_conn(struct conn *c)
{
	struct conn **pp;
	struct gpu_state *g = &gpus[c->gpu];

	epoll_ctl(epoll_fd, EPOLL_CTL_DEL, c->fd, NULL);
	close(c->fd);

	if (c->registered) {
		total_evictions++;
		log_info("client %016" PRIx64 " (%s/%s) disconnected",
			 c->id, c->pod_namespace[0] ? c->pod_namespace : "-",
			 c->pod_name[0] ? c->pod_name : "-");
	}
	if (c->wants_lock)
		queue_remove(g, c);
	if (g->lock_holder == c) {
		g->lock_held = 0;
		g->lock_holder = NULL;
		g->drop_lock_sent = 0;
	}
	for (pp = &clients; *pp != NULL; pp = &(*pp)->next) {
		if (*pp == c) {
			*pp = c->next;
			break;
		}
	}
	free(c);
	try_schedule(g);
}

static long total_mem_mib(void)
{
	long n = 0;
	struct conn *c;

	for (c = clients; c != NULL; c = c->next)
		if (c->registered)
			n += c->mem_mib;
	return n;
}

static void handle_status_req(struct conn *c)
{
	char big[64], buf[NVS_MSG_DATA_LEN];

	/* data: "<on>,<tq>,<nclients>,<qlen>[,<mem_mib>]" — the wire
	 * field is 20 bytes; mem_mib is last so only it can truncate. */
	snprintf(big, sizeof(big), "%d,%d,%d,%d,%ld", scheduler_on,
		 tq_seconds, client_count(), queue_len_all(),
		 total_mem_mib());
	nvs_strlcpy(buf, big, sizeof(buf));
	/* Through send_to so a failed/partial write marks the peer dead
	 * and the sweep evicts it — a torn frame on a kept-alive fd
	 * would misframe every later send to that peer. */
	send_to(c, NVS_STATUS, buf);
}

static int parse_gpu_index(const struct nvs_msg *m)
{
	char buf[NVS_MSG_DATA_LEN];
	long v;

	memcpy(buf, m->data, NVS_MSG_DATA_LEN);
	buf[NVS_MSG_DATA_LEN - 1] = '\0';
	if (strncmp(buf, "gpu", 3) != 0)
		return 0;
	v = strtol(buf + 3, NULL, 10);
	if (v < 0 || v >= MAX_GPUS)
		return 0;
	return (int)v;
}

static void process_msg(struct conn *c, const struct nvs_msg *m)
{
	struct gpu_state *g = &gpus[c->gpu];

	log_debug("recv %s from fd=%d id=%016" PRIx64,
		  nvs_msg_type_str(m->type), c->fd, c->id);

	switch (m->type) {
	case NVS_REGISTER: {
		char idbuf[NVS_MSG_DATA_LEN];

		if (c->registered) {
			log_warn("duplicate REGISTER from %016" PRIx64, c->id);
			break;
		}
		c->registered = 1;
		c->id = nvs_gen_id();
		c->gpu = parse_gpu_index(m);
		memcpy(c->pod_name, m->pod_name, NVS_POD_NAME_LEN);
		c->pod_name[NVS_POD_NAME_LEN - 1] = '\0';
		memcpy(c->pod_namespace, m->pod_namespace, NVS_POD_NS_LEN);
		c->pod_namespace[NVS_POD_NS_LEN - 1] = '\0';
		snprintf(idbuf, sizeof(idbuf), "%016" PRIx64, c->id);
		log_info("registered client %016" PRIx64 " (%s/%s) on gpu%d",
			 c->id,
			 c->pod_namespace[0] ? c->pod_namespace : "-",
			 c->pod_name[0] ? c->pod_name : "-", c->gpu);
		send_to(c, scheduler_on ? NVS_SCHED_ON : NVS_SCHED_OFF,
			idbuf);
		break;
	}
	case NVS_REQ_LOCK:
		if (!c->registered) {
			log_warn("REQ_LOCK from unregistered fd=%d", c->fd);
			break;
		}
		if (!scheduler_on)
			break; /* clients free-run while scheduling is off */
		if (g->lock_holder == c) {
			/* The holder re-requests between dropping
			 * own_lock and LOCK_RELEASED arriving: queue it
			 * again once the release is processed. */
			c->relock_pending = 1;
			break;
		}
		if (c->wants_lock)
			break;
		queue_push(g, c);
		if (!g->lock_held)
			try_schedule(g);
		else
			pthread_cond_broadcast(&timer_cv); /* arm preempt */
		break;
	case NVS_LOCK_RELEASED:
		if (g->lock_holder != c) {
			/* Late/duplicate release (early-release racing a
			 * preemption) — ignore. */
			log_debug("stale LOCK_RELEASED from %016" PRIx64,
				  c->id);
			break;
		}
		queue_remove(g, c);
		g->lock_held = 0;
		g->lock_holder = NULL;
		g->drop_lock_sent = 0;
		if (c->relock_pending) {
			c->relock_pending = 0;
			queue_push(g, c);
		}
		try_schedule(g);
		break;
	case NVS_SET_TQ: {
		char buf[NVS_MSG_DATA_LEN];
		long v;
		char *end = NULL;
		int gpu = -1;
		const char *num = buf;

		memcpy(buf, m->data, NVS_MSG_DATA_LEN);
		buf[NVS_MSG_DATA_LEN - 1] = '\0';
		/* "N" sets the global TQ; "gpuK:N" overrides one GPU. */
		if (strncmp(buf, "gpu", 3) == 0) {
			char *colon = strchr(buf, ':');

			if (colon == NULL) {
				log_warn("SET_TQ: malformed '%s'", buf);
				break;
			}
			gpu = (int)strtol(buf + 3, NULL, 10);
			if (gpu < 0 || gpu >= MAX_GPUS) {
				log_warn("SET_TQ: bad gpu in '%s'", buf);
				break;
			}
			num = colon + 1;
		}
		v = strtol(num, &end, 10);
		if (end == num || v < 1 || v > 86400) {
			log_warn("SET_TQ: invalid value '%s'", buf);
			break;
		}
		if (gpu >= 0) {
			gpus[gpu].tq_override = (int)v;
			log_info("TQ for gpu%d set to %d s", gpu, (int)v);
		} else {
			tq_seconds = (int)v;
			log_info("TQ set to %d s", tq_seconds);
		}
		pthread_cond_broadcast(&timer_cv);
		break;
	}
	case NVS_SCHED_ON:
		if (!scheduler_on) {
			int i;

			scheduler_on = 1;
			log_info("scheduling enabled");
			bcast_status();
			for (i = 0; i < MAX_GPUS; i++)
				try_schedule(&gpus[i]);
		}
		break;
	case NVS_SCHED_OFF:
		if (scheduler_on) {
			int i;

			scheduler_on = 0;
			log_info("scheduling disabled (free-for-all)");
			bcast_status();
			for (i = 0; i < MAX_GPUS; i++) {
				struct gpu_state *gi = &gpus[i];
				struct conn *it, *nx;

				for (it = gi->queue_head; it != NULL;
				     it = nx) {
					nx = it->qnext;
					it->qnext = NULL;
					it->wants_lock = 0;
				}
				gi->queue_head = gi->queue_tail = NULL;
				gi->lock_held = 0;
				gi->lock_holder = NULL;
				gi->drop_lock_sent = 0;
			}
		}
		break;
	case NVS_MEM_UPDATE: {
		char buf[NVS_MSG_DATA_LEN];
		char *end = NULL;
		long v;

		memcpy(buf, m->data, NVS_MSG_DATA_LEN);
		buf[NVS_MSG_DATA_LEN - 1] = '\0';
		/* "mem_mib[,cap_mib]" — cap feeds the global pressure
		 * policy (see gpu_pressure). */
		v = strtol(buf, &end, 10);
		if (c->registered && v >= 0)
			c->mem_mib = v;
		if (c->registered && end != NULL && *end == ',') {
			long cap = strtol(end + 1, NULL, 10);

			if (cap > 0)
				c->cap_mib = cap;
		}
		break;
	}
	case NVS_STATUS_REQ:
		handle_status_req(c);
		break;
	default:
		log_warn("unknown message type %u from fd=%d", m->type,
			 c->fd);
		break;
	}
}

/* Drain readable bytes from a connection; returns -1 when it died. */
static int handle_readable(struct conn *c)
{
	for (;;) {
		char *base = (char *)&c->inmsg;
		ssize_t n = read(c->fd, base + c->got,
				 NVS_MSG_SIZE - c->got);

		if (n == 0)
			return -1;
		if (n < 0) {
			if (errno == EINTR)
				continue;
			if (errno == EAGAIN || errno == EWOULDBLOCK)
				return 0;
			return -1;
		}
		c->got += (size_t)n;
		if (c->got == NVS_MSG_SIZE) {
			c->got = 0;
			process_msg(c, &c->inmsg);
			if (c->dead)
				return 0; /* stop; sweep evicts it */
		}
	}
}

static void dump_state(void)
{
	struct conn *c;
	int i;

	log_info("=== scheduler state ===");
	log_info("on=%d tq=%ds clients=%d evictions=%lu", scheduler_on,
		 tq_seconds, client_count(), total_evictions);
	for (i = 0; i < MAX_GPUS; i++) {
		struct gpu_state *g = &gpus[i];

		if (!g->grants && !queue_len(g))
			continue;
		log_info(" gpu%d: lock_held=%d holder=%016" PRIx64
			 " round=%lu grants=%lu preempts=%lu queue=%d "
			 "pressure=%d", i, g->lock_held,
			 g->lock_holder ? g->lock_holder->id : 0,
			 g->round, g->grants, g->preemptions,
			 queue_len(g), gpu_pressure(g));
	}
	for (c = clients; c != NULL; c = c->next)
		log_info("  conn fd=%d reg=%d id=%016" PRIx64 " gpu=%d "
			 "queued=%d mem=%ldMiB%s", c->fd, c->registered,
			 c->id, c->gpu, c->wants_lock, c->mem_mib,
			 gpus[c->gpu].lock_holder == c ? " [HOLDER]" : "");
}

static void on_sigusr1(int sig)
{
	(void)sig;
	dump_requested = 1;
}

int main(void)
{
	int lsock;
	pthread_t timer_tid;
	struct epoll_event ev, events[MAX_EPOLL_EVENTS];
	const char *dir;
	char dirbuf[NVS_SOCK_PATH_MAX];
	struct sigaction sa;

	nvs_log_init();
	signal(SIGPIPE, SIG_IGN);
	memset(&sa, 0, sizeof(sa));
	sa.sa_handler = on_sigusr1;
	sigaction(SIGUSR1, &sa, NULL);

	tq_seconds = (int)nvs_env_long("NVSHARE_TQ", NVS_DEFAULT_TQ, 1,
				       86400);
	scheduler_on = !nvs_env_bool("NVSHARE_SCHED_OFF", 0);

	/* Ensure the socket directory exists. */
	dir = getenv(NVS_SOCK_DIR_ENV);
	if (dir == NULL || dir[0] == '\0')
		dir = NVS_SOCK_DIR_DEFAULT;
	nvs_strlcpy(dirbuf, dir, sizeof(dirbuf));
	mkdir(dirbuf, 0777);

	true_or_exit(nvs_scheduler_path(sock_path) == 0);
	lsock = nvs_bind_listen(sock_path);
	if (lsock < 0)
		log_fatal("cannot bind %s: %s", sock_path, strerror(errno));

	epoll_fd = epoll_create1(EPOLL_CLOEXEC);
	true_or_exit(epoll_fd >= 0);
	memset(&ev, 0, sizeof(ev));
	ev.events = EPOLLIN;
	ev.data.ptr = NULL; /* NULL => listen socket */
	true_or_exit(epoll_ctl(epoll_fd, EPOLL_CTL_ADD, lsock, &ev) == 0);

	true_or_exit(pthread_create(&timer_tid, NULL, timer_thread, NULL)
		     == 0);

	log_info("nvshare-scheduler (amd) listening on %s (tq=%ds, "
		 "scheduling %s, up to %d GPUs)", sock_path, tq_seconds,
		 scheduler_on ? "on" : "off", MAX_GPUS);

	for (;;) {
		int nev, i;

		nev = epoll_wait(epoll_fd, events, MAX_EPOLL_EVENTS, 1000);
		if (nev < 0) {
			if (errno == EINTR) {
				if (dump_requested) {
					dump_requested = 0;
					pthread_mutex_lock(&g_mutex);
					dump_state();
					pthread_mutex_unlock(&g_mutex);
				}
				continue;
			}
			log_fatal("epoll_wait: %s", strerror(errno));
		}
		if (dump_requested) {
			dump_requested = 0;
			pthread_mutex_lock(&g_mutex);
			dump_state();
			pthread_mutex_unlock(&g_mutex);
		}
		pthread_mutex_lock(&g_mutex);
		for (i = 0; i < nev; i++) {
			struct conn *c = events[i].data.ptr;

			if (c == NULL) {
				/* New connection(s). */
				for (;;) {
					int fd = nvs_accept(lsock);
					struct conn *nc;

					if (fd < 0)
						break;
					{
						int fl = fcntl(fd, F_GETFL);
						fcntl(fd, F_SETFL,
						      fl | O_NONBLOCK);
					}
					nc = calloc(1, sizeof(*nc));
					true_or_exit(nc != NULL);
					nc->fd = fd;
					nc->next = clients;
					clients = nc;
					memset(&ev, 0, sizeof(ev));
					ev.events = EPOLLIN | EPOLLRDHUP;
					ev.data.ptr = nc;
					true_or_exit(epoll_ctl(epoll_fd,
						EPOLL_CTL_ADD, fd, &ev) == 0);
					log_debug("accepted fd=%d", fd);
				}
				continue;
			}
			if (events[i].events & (EPOLLERR | EPOLLHUP |
						EPOLLRDHUP)) {
				/* Drain anything pending, then evict. */
				handle_readable(c);
				delete_conn(c);
				continue;
			}
			if (events[i].events & EPOLLIN) {
				if (handle_readable(c) < 0)
					delete_conn(c);
			}
		}
		/* Sweep connections whose sends failed this pass. */
		{
			struct conn *c;
			int again = 1;

			while (again) {
				again = 0;
				for (c = clients; c != NULL; c = c->next) {
					if (c->dead) {
						delete_conn(c);
						again = 1;
						break;
					}
				}
			}
		}
		pthread_mutex_unlock(&g_mutex);
	}
	return 0;
}
