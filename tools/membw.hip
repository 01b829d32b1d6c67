// membw — measured achievable HBM read bandwidth on THIS box (streaming
// dwordx4 reads, grid-strided, result sunk to defeat DCE). Anchors the
// roofline "achievable" figure quoted in DESIGN.md.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>

__global__ void read_kernel(const ulonglong2 *__restrict__ p, size_t n,
                            unsigned long long *sink)
{
    unsigned long long acc = 0;
    const size_t stride = (size_t)gridDim.x * blockDim.x;
    for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        const ulonglong2 v = p[i];
        acc ^= v.x ^ v.y;
    }
    if (acc == 0xDEADBEEFCAFEF00DULL) atomicAdd(sink, acc);
}

int main()
{
    const size_t bytes = 8ull << 30;               // 8 GiB
    const size_t n = bytes / 16;
    ulonglong2 *d = nullptr;
    unsigned long long *sink = nullptr;
    if (hipMalloc(&d, bytes) != hipSuccess ||
        hipMalloc(&sink, 8) != hipSuccess ||
        hipMemset(d, 0x5A, bytes) != hipSuccess) {
        fprintf(stderr, "membw: allocation failed\n");
        return 1;
    }
    hipEvent_t e0, e1;
    hipEventCreate(&e0);
    hipEventCreate(&e1);
    for (int rep = 0; rep < 3; rep++) {
        hipEventRecord(e0);
        read_kernel<<<dim3(16384), dim3(256)>>>(d, n, sink);
        hipEventRecord(e1);
        hipEventSynchronize(e1);
        float ms;
        hipEventElapsedTime(&ms, e0, e1);
        printf("read %zu GiB in %.3f ms = %.2f TB/s\n",
               bytes >> 30, ms, bytes / (ms / 1e3) / 1e12);
    }
    return 0;
}
