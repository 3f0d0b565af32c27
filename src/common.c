/* nvshare-amd common substrate implementation. */
#include "common.h"

int nvs_debug_enabled = 0;

void nvs_log_init(void)
{
	const char *v = getenv("NVSHARE_DEBUG");
	nvs_debug_enabled = (v != NULL && v[0] != '\0' && v[0] != '0');
}

ssize_t nvs_write_whole(int fd, const void *buf, size_t count)
{
	const char *p = buf;
	size_t left = count;
	ssize_t n;

	while (left > 0) {
		RETRY_EINTR(n, write(fd, p, left));
		if (n <= 0)
			return -1;
		p += n;
		left -= (size_t)n;
	}
	return (ssize_t)count;
}

ssize_t nvs_read_whole(int fd, void *buf, size_t count)
{
	char *p = buf;
	size_t left = count;
	ssize_t n;

	while (left > 0) {
		RETRY_EINTR(n, read(fd, p, left));
		if (n == 0)
			return -1; /* EOF mid-message */
		if (n < 0)
			return -1;
		p += n;
		left -= (size_t)n;
	}
	return (ssize_t)count;
}

size_t nvs_strlcpy(char *dst, const char *src, size_t size)
{
	size_t srclen = strlen(src);

	if (size > 0) {
		size_t n = (srclen >= size) ? size - 1 : srclen;
		memcpy(dst, src, n);
		dst[n] = '\0';
	}
	return srclen;
}

long nvs_env_long(const char *name, long dflt, long lo, long hi)
{
	const char *v = getenv(name);
	char *end = NULL;
	long r;

	if (v == NULL || v[0] == '\0')
		return dflt;
	r = strtol(v, &end, 10);
	if (end == v)
		return dflt;
	if (r < lo)
		r = lo;
	if (r > hi)
		r = hi;
	return r;
}

int nvs_env_bool(const char *name, int dflt)
{
	const char *v = getenv(name);

	if (v == NULL)
		return dflt;
	return (v[0] != '\0' && v[0] != '0');
}
