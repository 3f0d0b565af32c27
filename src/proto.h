/*
 * nvshare-amd — control-plane wire protocol over Unix-domain sockets.
 *
 * Wire-compatible with the reference protocol (/root/reference/src/comm.h:
 * 537-byte packed message, enum values 1..8, socket at
 * <sock_dir>/scheduler.sock).  Types 9..10 are nvshare-amd extensions
 * (status query) that old peers never send and the daemon only answers
 * when asked, so compatibility is a strict superset.
 */
#ifndef NVSHARE_PROTO_H
#define NVSHARE_PROTO_H

#include <stdint.h>
#include <sys/un.h>

#define NVS_POD_NAME_LEN 254
#define NVS_POD_NS_LEN   254
#define NVS_MSG_DATA_LEN 20

#define NVS_SOCK_DIR_DEFAULT "/var/run/nvshare/"
#define NVS_SOCK_DIR_ENV     "NVSHARE_SOCK_DIR"
#define NVS_SCHED_SOCK_NAME  "scheduler.sock"
#define NVS_SOCK_PATH_MAX    sizeof(((struct sockaddr_un *)0)->sun_path)

enum nvs_msg_type {
	NVS_REGISTER      = 1,
	NVS_SCHED_ON      = 2,
	NVS_SCHED_OFF     = 3,
	NVS_REQ_LOCK      = 4,
	NVS_LOCK_OK       = 5,
	NVS_DROP_LOCK     = 6,
	NVS_LOCK_RELEASED = 7,
	NVS_SET_TQ        = 8,
	/* nvshare-amd extensions */
	NVS_STATUS_REQ    = 9,
	NVS_STATUS        = 10,
	NVS_MEM_UPDATE    = 11,  /* client -> scheduler: tracked MiB */
};

#define NVS_MSG_TYPE_MAX NVS_MEM_UPDATE

struct nvs_msg {
	uint8_t type;
	char pod_name[NVS_POD_NAME_LEN];
	char pod_namespace[NVS_POD_NS_LEN];
	uint64_t id;
	char data[NVS_MSG_DATA_LEN];
} __attribute__((__packed__));

#define NVS_MSG_SIZE 537
_Static_assert(sizeof(struct nvs_msg) == NVS_MSG_SIZE,
	       "wire message must be 537 bytes");

const char *nvs_msg_type_str(uint8_t type);

/* 64-bit random id (non-zero). */
uint64_t nvs_gen_id(void);

/* Resolve the scheduler socket path into buf (size NVS_SOCK_PATH_MAX).
 * Honors NVSHARE_SOCK_DIR. Returns 0, or -1 if the path would overflow. */
int nvs_scheduler_path(char *buf);

/* Create/bind/listen a Unix stream socket at path (unlinks stale file,
 * chmods 0777 so unprivileged clients can connect). Returns fd or -1. */
int nvs_bind_listen(const char *path);

/* Connect (blocking) to a Unix stream socket. Returns fd or -1. */
int nvs_connect_path(const char *path);

/* accept4 with CLOEXEC; returns fd or -1 (EAGAIN => -1 with errno). */
int nvs_accept(int lsock);

/* Blocking full-message send/recv; return 0 or -1. */
int nvs_send_msg(int fd, const struct nvs_msg *m);
int nvs_recv_msg(int fd, struct nvs_msg *m);

/* Fill a message: zeroes, sets type/id and copies data string if given. */
void nvs_msg_init(struct nvs_msg *m, uint8_t type, uint64_t id,
		  const char *data);

#endif /* NVSHARE_PROTO_H */
