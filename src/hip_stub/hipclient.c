/*
 * hipclient — scripted HIP workload for hook/scheduler tests (CPU, via
 * the stub libamdhip64).  Run under LD_PRELOAD=libnvshare.so to drive
 * the full interposer + scheduler stack without a GPU.
 *
 * Usage: hipclient [--allocs N] [--alloc-mib M] [--iters K]
 *                  [--copy-bytes B] [--sleep-us U] [--sync-every S]
 * Prints "PASS <seconds>" on success, "OOM" and exit 3 when an
 * allocation is rejected.
 */
#define _GNU_SOURCE
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <time.h>
#include <unistd.h>

typedef int hipError_t;
typedef void *hipStream_t;
typedef struct { unsigned x, y, z; } dim3_t;

extern hipError_t hipMalloc(void **, size_t);
extern hipError_t hipFree(void *);
extern hipError_t hipMemGetInfo(size_t *, size_t *);
extern hipError_t hipLaunchKernel(const void *, dim3_t, dim3_t, void **,
				  size_t, hipStream_t);
extern hipError_t hipMemcpyAsync(void *, const void *, size_t, int,
				 hipStream_t);
extern hipError_t hipDeviceSynchronize(void);

static double now_s(void)
{
	struct timespec ts;
	clock_gettime(CLOCK_MONOTONIC, &ts);
	return (double)ts.tv_sec + (double)ts.tv_nsec / 1e9;
}

int main(int argc, char **argv)
{
	long allocs = 1, alloc_mib = 64, iters = 100, copy_bytes = 4096;
	long sleep_us = 0, sync_every = 0;
	void **bufs;
	char *host;
	long i, k;
	double t0;
	dim3_t grid = { 1, 1, 1 }, block = { 64, 1, 1 };

	for (i = 1; i < argc - 1; i += 2) {
		if (strcmp(argv[i], "--allocs") == 0)
			allocs = atol(argv[i + 1]);
		else if (strcmp(argv[i], "--alloc-mib") == 0)
			alloc_mib = atol(argv[i + 1]);
		else if (strcmp(argv[i], "--iters") == 0)
			iters = atol(argv[i + 1]);
		else if (strcmp(argv[i], "--copy-bytes") == 0)
			copy_bytes = atol(argv[i + 1]);
		else if (strcmp(argv[i], "--sleep-us") == 0)
			sleep_us = atol(argv[i + 1]);
		else if (strcmp(argv[i], "--sync-every") == 0)
			sync_every = atol(argv[i + 1]);
		else {
			fprintf(stderr, "unknown arg %s\n", argv[i]);
			return 2;
		}
	}

	t0 = now_s();
	bufs = calloc((size_t)allocs, sizeof(void *));
	host = calloc(1, (size_t)copy_bytes);
	if (bufs == NULL || host == NULL)
		return 2;

	for (i = 0; i < allocs; i++) {
		hipError_t r = hipMalloc(&bufs[i],
					 (size_t)alloc_mib * 1024 * 1024);
		if (r != 0) {
			printf("OOM\n");
			fflush(stdout);
			return 3;
		}
	}

	{
		size_t freeb = 0, totalb = 0;

		hipMemGetInfo(&freeb, &totalb);
		fprintf(stderr, "hipclient: free=%zu MiB total=%zu MiB\n",
			freeb >> 20, totalb >> 20);
	}

	for (k = 0; k < iters; k++) {
		if (hipLaunchKernel((void *)0xdead, grid, block, NULL, 0,
				    NULL) != 0)
			return 4;
		if (copy_bytes > 0 &&
		    hipMemcpyAsync(bufs[k % allocs], host,
				   (size_t)copy_bytes, 1, NULL) != 0)
			return 5;
		if (sync_every > 0 && (k + 1) % sync_every == 0)
			hipDeviceSynchronize();
		if (sleep_us > 0)
			usleep((useconds_t)sleep_us);
	}
	hipDeviceSynchronize();

	for (i = 0; i < allocs; i++)
		hipFree(bufs[i]);

	printf("PASS %.3f\n", now_s() - t0);
	fflush(stdout);
	return 0;
}
