/*
 * nvshare-amd gfx950 utility kernels (libnvshare_hiputil.so).
 *
 * Small, purposeful device code for an otherwise host-side system:
 *   - touch_pages:  fault/migrate a managed buffer from the GPU at a
 *                   given stride; the probe behind faultbench (measures
 *                   gfx950 XNACK/HMM demand-paging throughput, the #1
 *                   porting risk called out in SURVEY.md §7).
 *   - busy_kernel:  spins for a requested duration; emulates clients
 *                   with controlled duty cycles in GPU tests.
 *   - stream_triad: HBM bandwidth probe (a[i] = b[i] + s*c[i]) used to
 *                   quantify post-migration steady-state bandwidth vs
 *                   the ~6.3 TB/s device ceiling.
 *
 * The grid sizing follows the MI355X occupancy rules: 256 CUs across
 * 8 XCDs want >>256 workgroups in flight (see
 * /opt/skills/guides/MI355X_MICROARCH.md, chip-level parameters).
 */
#include <hip/hip_runtime.h>

#include <cstdint>

#define CHECK(x)                                                           \
	do {                                                               \
		hipError_t err_ = (x);                                     \
		if (err_ != hipSuccess)                                    \
			return (int)err_;                                  \
	} while (0)

__global__ void touch_pages_kernel(float *buf, size_t n, size_t stride,
				   float val)
{
	size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
	size_t nthreads = (size_t)gridDim.x * blockDim.x;

	for (size_t idx = i * stride; idx < n; idx += nthreads * stride)
		buf[idx] += val;
}

__global__ void read_pages_kernel(const float *buf, size_t n,
				  size_t stride, float *out)
{
	size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
	size_t nthreads = (size_t)gridDim.x * blockDim.x;
	float acc = 0.f;

	for (size_t idx = i * stride; idx < n; idx += nthreads * stride)
		acc += buf[idx];
	if (acc == -1.f) /* never true; defeats DCE */
		out[0] = acc;
}

__global__ void busy_kernel(int64_t cycles)
{
	int64_t start = wall_clock64();

	while (wall_clock64() - start < cycles) {
		/* spin; one wave per CU is enough to look "busy" */
	}
}

__global__ void stream_triad_kernel(float *__restrict__ a,
				    const float *__restrict__ b,
				    const float *__restrict__ c, float s,
				    size_t n)
{
	size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
	size_t nthreads = (size_t)gridDim.x * blockDim.x;

	for (size_t idx = i * 4; idx + 3 < n; idx += nthreads * 4) {
		float4 vb = *reinterpret_cast<const float4 *>(&b[idx]);
		float4 vc = *reinterpret_cast<const float4 *>(&c[idx]);
		float4 va = { vb.x + s * vc.x, vb.y + s * vc.y,
			      vb.z + s * vc.z, vb.w + s * vc.w };
		*reinterpret_cast<float4 *>(&a[idx]) = va;
	}
}

__global__ void tiny_kernel(float *p)
{
	if (threadIdx.x == 0 && blockIdx.x == 0)
		p[0] += 1.0f;
}

extern "C" {

/* Launch n trivial kernels touching p; measures per-launch overhead
 * (managed vs plain pointers — isolates the launch-dense workload
 * penalty of the managed conversion). */
int nvs_launch_burst(float *p, int n)
{
	for (int i = 0; i < n; i++)
		hipLaunchKernelGGL(tiny_kernel, dim3(1), dim3(64), 0,
				   nullptr, p);
	CHECK(hipGetLastError());
	CHECK(hipDeviceSynchronize());
	return 0;
}

/* Touch every `stride`-th float of buf[0..n) from the GPU. */
int nvs_touch_pages(float *buf, size_t n, size_t stride, float val,
		    void *stream)
{
	int blocks = 2048; /* >> 256 CUs, fills all 8 XCDs */

	if (stride == 0)
		stride = 1;
	hipLaunchKernelGGL(touch_pages_kernel, dim3(blocks), dim3(256), 0,
			   (hipStream_t)stream, buf, n, stride, val);
	return (int)hipGetLastError();
}

int nvs_read_pages(const float *buf, size_t n, size_t stride, float *out,
		   void *stream)
{
	int blocks = 2048;

	if (stride == 0)
		stride = 1;
	hipLaunchKernelGGL(read_pages_kernel, dim3(blocks), dim3(256), 0,
			   (hipStream_t)stream, buf, n, stride, out);
	return (int)hipGetLastError();
}

/* Spin all CUs for ~ms milliseconds (wall clock ~100 MHz). */
int nvs_busy(double ms, void *stream)
{
	/* wall_clock64 ticks at a fixed 100 MHz on CDNA. */
	int64_t cycles = (int64_t)(ms * 1e5);

	hipLaunchKernelGGL(busy_kernel, dim3(256), dim3(64), 0,
			   (hipStream_t)stream, cycles);
	return (int)hipGetLastError();
}

int nvs_stream_triad(float *a, const float *b, const float *c, float s,
		     size_t n, void *stream)
{
	hipLaunchKernelGGL(stream_triad_kernel, dim3(4096), dim3(256), 0,
			   (hipStream_t)stream, a, b, c, s, n);
	return (int)hipGetLastError();
}

int nvs_hip_malloc_managed(void **p, size_t bytes)
{
	CHECK(hipMallocManaged(p, bytes, hipMemAttachGlobal));
	return 0;
}

int nvs_hip_malloc(void **p, size_t bytes)
{
	CHECK(hipMalloc(p, bytes));
	return 0;
}

int nvs_hip_free(void *p)
{
	CHECK(hipFree(p));
	return 0;
}

int nvs_hip_prefetch(const void *p, size_t bytes, int device)
{
	CHECK(hipMemPrefetchAsync(p, bytes, device, nullptr));
	return 0;
}

/* advice: 3=SetPreferredLocation, 5=SetAccessedBy, 100=SetCoarseGrain */
int nvs_hip_advise(const void *p, size_t bytes, int advice, int device)
{
	CHECK(hipMemAdvise(p, bytes, (hipMemoryAdvise)advice, device));
	return 0;
}

/* Chunked prefetch across n_streams concurrent streams (probes SDMA
 * parallelism for post-preemption working-set restore). */
int nvs_hip_prefetch_chunked(const void *p, size_t bytes, int device,
			     int n_streams, size_t chunk_bytes)
{
	hipStream_t streams[16];
	int i, n = n_streams;

	if (n < 1)
		n = 1;
	if (n > 16)
		n = 16;
	if (chunk_bytes == 0)
		chunk_bytes = 256ULL << 20;
	for (i = 0; i < n; i++)
		CHECK(hipStreamCreateWithFlags(&streams[i],
					       hipStreamNonBlocking));
	{
		const char *base = (const char *)p;
		size_t off = 0;
		int s = 0;

		while (off < bytes) {
			size_t len = bytes - off < chunk_bytes ?
				     bytes - off : chunk_bytes;
			hipError_t e = hipMemPrefetchAsync(
				base + off, len, device, streams[s]);
			if (e != hipSuccess) {
				for (i = 0; i < n; i++)
					(void)hipStreamDestroy(
						streams[i]);
				return (int)e;
			}
			off += len;
			s = (s + 1) % n;
		}
	}
	for (i = 0; i < n; i++) {
		hipError_t e = hipStreamSynchronize(streams[i]);

		(void)hipStreamDestroy(streams[i]);
		if (e != hipSuccess)
			return (int)e;
	}
	return 0;
}

int nvs_hip_sync(void)
{
	CHECK(hipDeviceSynchronize());
	return 0;
}

int nvs_hip_mem_get_info(size_t *free_b, size_t *total_b)
{
	CHECK(hipMemGetInfo(free_b, total_b));
	return 0;
}

} /* extern "C" */
