/* nvshare-amd wire protocol implementation. */
#define _GNU_SOURCE
#include <fcntl.h>
#include <poll.h>
#include <stdio.h>
#include <string.h>
#include <sys/socket.h>
#include <sys/stat.h>
#include <sys/types.h>
#include <sys/un.h>
#include <unistd.h>

#include "common.h"
#include "proto.h"

static const char *type_names[] = {
	[0] = "INVALID",
	[NVS_REGISTER] = "REGISTER",
	[NVS_SCHED_ON] = "SCHED_ON",
	[NVS_SCHED_OFF] = "SCHED_OFF",
	[NVS_REQ_LOCK] = "REQ_LOCK",
	[NVS_LOCK_OK] = "LOCK_OK",
	[NVS_DROP_LOCK] = "DROP_LOCK",
	[NVS_LOCK_RELEASED] = "LOCK_RELEASED",
	[NVS_SET_TQ] = "SET_TQ",
	[NVS_STATUS_REQ] = "STATUS_REQ",
	[NVS_STATUS] = "STATUS",
	[NVS_MEM_UPDATE] = "MEM_UPDATE",
};

const char *nvs_msg_type_str(uint8_t type)
{
	if (type > NVS_MSG_TYPE_MAX)
		return "UNKNOWN";
	return type_names[type];
}

uint64_t nvs_gen_id(void)
{
	uint64_t id = 0;
	int fd, i;
	ssize_t n;

	fd = open("/dev/urandom", O_RDONLY | O_CLOEXEC);
	if (fd >= 0) {
		n = nvs_read_whole(fd, &id, sizeof(id));
		close(fd);
		if (n == sizeof(id) && id != 0)
			return id;
	}
	/* Fallback: clock + pid mix. */
	id = (uint64_t)nvs_now_ns();
	id ^= (uint64_t)getpid() << 32;
	for (i = 0; id == 0 && i < 4; i++)
		id = (uint64_t)nvs_now_ns() + 1;
	return id;
}

int nvs_scheduler_path(char *buf)
{
	const char *dir = getenv(NVS_SOCK_DIR_ENV);
	int n;

	if (dir == NULL || dir[0] == '\0')
		dir = NVS_SOCK_DIR_DEFAULT;
	n = snprintf(buf, NVS_SOCK_PATH_MAX, "%s%s%s", dir,
		     dir[strlen(dir) - 1] == '/' ? "" : "/",
		     NVS_SCHED_SOCK_NAME);
	if (n < 0 || (size_t)n >= NVS_SOCK_PATH_MAX)
		return -1;
	return 0;
}

int nvs_bind_listen(const char *path)
{
	struct sockaddr_un addr;
	int fd;

	if (strlen(path) >= NVS_SOCK_PATH_MAX)
		return -1;

	fd = socket(AF_UNIX, SOCK_STREAM | SOCK_CLOEXEC | SOCK_NONBLOCK, 0);
	if (fd < 0)
		return -1;

	unlink(path);
	memset(&addr, 0, sizeof(addr));
	addr.sun_family = AF_UNIX;
	nvs_strlcpy(addr.sun_path, path, sizeof(addr.sun_path));
	if (bind(fd, (struct sockaddr *)&addr, sizeof(addr)) < 0) {
		close(fd);
		return -1;
	}
	/* Clients may run as arbitrary users inside containers. */
	chmod(path, 0777);
	if (listen(fd, 128) < 0) {
		close(fd);
		return -1;
	}
	return fd;
}

int nvs_connect_path(const char *path)
{
	struct sockaddr_un addr;
	int fd, ret;

	if (strlen(path) >= NVS_SOCK_PATH_MAX)
		return -1;

	fd = socket(AF_UNIX, SOCK_STREAM | SOCK_CLOEXEC, 0);
	if (fd < 0)
		return -1;
	memset(&addr, 0, sizeof(addr));
	addr.sun_family = AF_UNIX;
	nvs_strlcpy(addr.sun_path, path, sizeof(addr.sun_path));
	RETRY_EINTR(ret, connect(fd, (struct sockaddr *)&addr, sizeof(addr)));
	if (ret < 0) {
		close(fd);
		return -1;
	}
	return fd;
}

int nvs_accept(int lsock)
{
	int fd;

	RETRY_EINTR(fd, accept4(lsock, NULL, NULL, SOCK_CLOEXEC));
	return fd;
}

/*
 * Frame-safe send that also works on nonblocking fds (the daemon keeps
 * client sockets nonblocking for reads): on EAGAIN, poll for
 * writability briefly (500 ms total) so a frame is never torn
 * mid-message.  The budget is short on purpose: the daemon sends while
 * holding its global lock, so a slow peer must not head-of-line block
 * other clients (the reference evicted instantly on partial sends; we
 * just add a small grace).  A peer that stays unwritable is broken and
 * the caller must treat the failure as fatal for the connection
 * (framing can no longer be trusted).
 */
int nvs_send_msg(int fd, const struct nvs_msg *m)
{
	const char *p = (const char *)m;
	size_t left = NVS_MSG_SIZE;
	int waited_ms = 0;

	while (left > 0) {
		ssize_t n;

		RETRY_EINTR(n, write(fd, p, left));
		if (n > 0) {
			p += n;
			left -= (size_t)n;
			continue;
		}
		if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
			struct pollfd pfd = { .fd = fd,
					      .events = POLLOUT };
			int pr;

			if (waited_ms >= 500)
				return -1;
			RETRY_EINTR(pr, poll(&pfd, 1, 50));
			waited_ms += 50;
			continue;
		}
		return -1;
	}
	return 0;
}

int nvs_recv_msg(int fd, struct nvs_msg *m)
{
	if (nvs_read_whole(fd, m, NVS_MSG_SIZE) != NVS_MSG_SIZE)
		return -1;
	return 0;
}

void nvs_msg_init(struct nvs_msg *m, uint8_t type, uint64_t id,
		  const char *data)
{
	memset(m, 0, sizeof(*m));
	m->type = type;
	m->id = id;
	if (data != NULL)
		nvs_strlcpy(m->data, data, sizeof(m->data));
}
