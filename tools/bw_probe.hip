/* bw_probe — measures achievable HBM read bandwidth on gfx950 for the
 * (build: hipcc --offload-arch=gfx950 -O3 tools/bw_probe.hip -o tools/bw_probe)
 * access shapes our scan kernels use: plain 8B grid-stride, 16B (uint4-
 * style) grid-stride, nontemporal 8B, and block-chunked 8B.  Prints
 * GB/s per (shape, grid) so kernel targets are set from MEASURED
 * ceilings, not the spec sheet. */
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>

#define CHK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
    fprintf(stderr, "HIP error %s @%d\n", hipGetErrorString(e), __LINE__); return 1; } } while (0)

__global__ void k_sum8(const uint64_t *p, int64_t n, uint64_t *out)
{
    int64_t i = blockIdx.x * (int64_t) blockDim.x + threadIdx.x;
    int64_t stride = gridDim.x * (int64_t) blockDim.x;
    uint64_t s = 0;
    for (; i < n; i += stride) s += p[i];
    if (s == 0xdeadbeefULL) *out = s;   /* keep the loads */
}

__global__ void k_sum8_nt(const uint64_t *p, int64_t n, uint64_t *out)
{
    int64_t i = blockIdx.x * (int64_t) blockDim.x + threadIdx.x;
    int64_t stride = gridDim.x * (int64_t) blockDim.x;
    uint64_t s = 0;
    for (; i < n; i += stride) s += __builtin_nontemporal_load(p + i);
    if (s == 0xdeadbeefULL) *out = s;
}

__global__ void k_sum16(const ulonglong2 *p, int64_t n2, uint64_t *out)
{
    int64_t i = blockIdx.x * (int64_t) blockDim.x + threadIdx.x;
    int64_t stride = gridDim.x * (int64_t) blockDim.x;
    uint64_t s = 0;
    for (; i < n2; i += stride) { ulonglong2 v = p[i]; s += v.x + v.y; }
    if (s == 0xdeadbeefULL) *out = s;
}

__global__ void k_sum32(const ulonglong4 *p, int64_t n4, uint64_t *out)
{
    int64_t i = blockIdx.x * (int64_t) blockDim.x + threadIdx.x;
    int64_t stride = gridDim.x * (int64_t) blockDim.x;
    uint64_t s = 0;
    for (; i < n4; i += stride) { ulonglong4 v = p[i]; s += v.x + v.y + v.z + v.w; }
    if (s == 0xdeadbeefULL) *out = s;
}

__global__ void k_sum8_chunk(const uint64_t *p, int64_t n, uint64_t *out)
{
    int64_t chunk = (n + gridDim.x - 1) / gridDim.x;
    int64_t lo = blockIdx.x * chunk;
    int64_t hi = min(lo + chunk, n);
    uint64_t s = 0;
    for (int64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) s += p[i];
    if (s == 0xdeadbeefULL) *out = s;
}

int main()
{
    const int64_t GB = 1LL << 30;
    const int64_t bytes = 16 * GB;
    const int64_t n = bytes / 8;
    uint64_t *d, *dout;
    CHK(hipMalloc(&d, bytes));
    CHK(hipMalloc(&dout, 8));
    CHK(hipMemset(d, 1, bytes));
    hipEvent_t e0, e1;
    CHK(hipEventCreate(&e0));
    CHK(hipEventCreate(&e1));
    int grids[] = {2048, 8192, 32768, 131072};
    auto run = [&](const char *name, auto kern, auto ptr, int64_t nn, int grid) {
        hipLaunchKernelGGL(kern, dim3(grid), dim3(256), 0, 0, ptr, nn, dout);
        (void) hipDeviceSynchronize();
        (void) hipEventRecord(e0);
        for (int r = 0; r < 3; r++)
            hipLaunchKernelGGL(kern, dim3(grid), dim3(256), 0, 0, ptr, nn, dout);
        (void) hipEventRecord(e1);
        (void) hipEventSynchronize(e1);
        float ms = 0;
        (void) hipEventElapsedTime(&ms, e0, e1);
        printf("%-14s grid %6d : %8.1f GB/s\n", name, grid,
               3.0 * bytes / (ms / 1e3) / 1e9);
    };
    for (int g : grids) run("sum8", k_sum8, (const uint64_t *) d, n, g);
    for (int g : grids) run("sum8_nt", k_sum8_nt, (const uint64_t *) d, n, g);
    for (int g : grids) run("sum16", k_sum16, (const ulonglong2 *) d, n / 2, g);
    for (int g : grids) run("sum32", k_sum32, (const ulonglong4 *) d, n / 4, g);
    for (int g : grids) run("sum8_chunk", k_sum8_chunk, (const uint64_t *) d, n, g);
    return 0;
}
