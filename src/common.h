/*
 * nvshare-amd — common substrate: logging, robust I/O, time helpers.
 *
 * MI355X-native rebuild of the capabilities of grgalex/nvshare.
 * Parity reference: /root/reference/src/common.{c,h} (log macro tiers,
 * read/write_whole loops, RETRY_INTR). Fresh implementation.
 */
#ifndef NVSHARE_COMMON_H
#define NVSHARE_COMMON_H

#include <errno.h>
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/types.h>
#include <time.h>
#include <unistd.h>

/* Set from NVSHARE_DEBUG at init (nvs_log_init). */
extern int nvs_debug_enabled;

void nvs_log_init(void);

#define nvs_log_raw(level, fmt, ...)                                        \
	do {                                                                \
		fprintf(stderr, "[NVSHARE][" level "] " fmt "\n",           \
			##__VA_ARGS__);                                     \
		fflush(stderr);                                             \
	} while (0)

#define log_info(fmt, ...)  nvs_log_raw("INFO", fmt, ##__VA_ARGS__)
#define log_warn(fmt, ...)  nvs_log_raw("WARN", fmt, ##__VA_ARGS__)
#define log_debug(fmt, ...)                                                 \
	do {                                                                \
		if (nvs_debug_enabled)                                      \
			nvs_log_raw("DEBUG", fmt, ##__VA_ARGS__);           \
	} while (0)
#define log_fatal(fmt, ...)                                                 \
	do {                                                                \
		nvs_log_raw("FATAL", fmt, ##__VA_ARGS__);                   \
		exit(1);                                                    \
	} while (0)

/* Evaluate expr; exit with a message when it is false. */
#define true_or_exit(expr)                                                  \
	do {                                                                \
		if (!(expr))                                                \
			log_fatal("%s:%d: check failed: %s (errno=%d %s)",  \
				  __FILE__, __LINE__, #expr, errno,         \
				  strerror(errno));                         \
	} while (0)

/* Retry a syscall while it returns -1/EINTR. */
#define RETRY_EINTR(ret, call)                                              \
	do {                                                                \
		ret = (call);                                               \
	} while (ret == -1 && errno == EINTR)

/* Write/read exactly count bytes on a blocking fd; -1 on error/EOF. */
ssize_t nvs_write_whole(int fd, const void *buf, size_t count);
ssize_t nvs_read_whole(int fd, void *buf, size_t count);

/* BSD strlcpy: bounded copy, always NUL-terminates (size > 0). */
size_t nvs_strlcpy(char *dst, const char *src, size_t size);

/* Monotonic clock in nanoseconds. */
static inline int64_t nvs_now_ns(void)
{
	struct timespec ts;
	clock_gettime(CLOCK_MONOTONIC, &ts);
	return (int64_t)ts.tv_sec * 1000000000LL + ts.tv_nsec;
}

static inline double nvs_now_s(void)
{
	return (double)nvs_now_ns() / 1e9;
}

/* Env var as long with default; clamps to [lo, hi]. */
long nvs_env_long(const char *name, long dflt, long lo, long hi);

/* Env var as boolean (unset/"0"/"" => 0, anything else => 1). */
int nvs_env_bool(const char *name, int dflt);

#define NVS_MIB ((size_t)1024 * 1024)

#endif /* NVSHARE_COMMON_H */
