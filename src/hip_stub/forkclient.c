/*
 * forkclient — fork-safety test client (CPU stub).
 * Parent registers with the scheduler and does GPU work; the forked
 * child must free-run (no scheduler protocol corruption, no deadlock).
 */
#define _GNU_SOURCE
#include <stdio.h>
#include <stdlib.h>
#include <sys/wait.h>
#include <unistd.h>

typedef int hipError_t;
typedef void *hipStream_t;
typedef struct { unsigned x, y, z; } dim3_t;

extern hipError_t hipMalloc(void **, size_t);
extern hipError_t hipLaunchKernel(const void *, dim3_t, dim3_t, void **,
				  size_t, hipStream_t);
extern hipError_t hipDeviceSynchronize(void);

int main(void)
{
	void *buf = NULL;
	dim3_t g = { 1, 1, 1 }, b = { 64, 1, 1 };
	pid_t pid;
	int status = 0, i;

	if (hipMalloc(&buf, 1 << 20) != 0)
		return 2;
	for (i = 0; i < 10; i++)
		if (hipLaunchKernel((void *)0x1, g, b, NULL, 0, NULL) != 0)
			return 3;

	pid = fork();
	if (pid == 0) {
		/* Child: must not hang or touch the parent's socket. */
		for (i = 0; i < 10; i++)
			if (hipLaunchKernel((void *)0x1, g, b, NULL, 0,
					    NULL) != 0)
				_exit(4);
		hipDeviceSynchronize();
		_exit(0);
	}
	for (i = 0; i < 10; i++)
		if (hipLaunchKernel((void *)0x1, g, b, NULL, 0, NULL) != 0)
			return 5;
	hipDeviceSynchronize();
	if (waitpid(pid, &status, 0) != pid || !WIFEXITED(status) ||
	    WEXITSTATUS(status) != 0) {
		printf("CHILD_FAILED %d\n", status);
		return 6;
	}
	printf("PASS\n");
	return 0;
}
