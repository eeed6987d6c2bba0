// Ablation probe for the bucket-count pass: what actually bounds
// 1e9-particle cell counting at 1024^3?  Variants:
//   read   : loads + cell computation only (no atomics) — read ceiling
//   atom   : + global atomicAdd(&counts[cell]) (the real kernel)
//   atomsm : + atomicAdd into a 4096-entry array (L2-resident, hot)
//   lds    : + atomicAdd into a 4096-entry LDS histogram
//   copy   : read + streaming compacted write (scatter write ceiling)
// Build: hipcc --offload-arch=gfx950 -O3 count_probe.hip -o count_probe
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#define CHECK(x) do { hipError_t e = (x); if (e) { \
    printf("ERR %s at %d\n", hipGetErrorString(e), __LINE__); exit(1); } \
} while (0)

__device__ __forceinline__ int64_t cell_of(double x, double y, double z,
                                           double inv, int64_t n) {
    const int64_t ix = ((int64_t)floor(x * inv)) & (n - 1);
    const int64_t iy = ((int64_t)floor(y * inv)) & (n - 1);
    const int64_t iz = ((int64_t)floor(z * inv)) & (n - 1);
    return (ix * n + iy) * n + iz;
}

template <int MODE>
__global__ void kprobe(const double* __restrict__ pos, int64_t n,
                       double inv, int64_t nm, int* __restrict__ counts,
                       int64_t* __restrict__ sink, double* __restrict__ out)
{
    __shared__ int lh[4096];
    if (MODE == 3)
        for (int i = threadIdx.x; i < 4096; i += blockDim.x) lh[i] = 0;
    if (MODE == 3) __syncthreads();
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int64_t acc = 0;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < n; i += stride) {
        const double x = pos[3 * i], y = pos[3 * i + 1],
                     z = pos[3 * i + 2];
        const int64_t c = cell_of(x, y, z, inv, nm);
        if (MODE == 0) acc += c;
        else if (MODE == 1) atomicAdd(&counts[c], 1);
        else if (MODE == 2) atomicAdd(&counts[c & 4095], 1);
        else if (MODE == 3) atomicAdd(&lh[c & 4095], 1);
        else if (MODE == 4) { out[i] = x + y + z; acc += c; }
    }
    if (MODE == 0 || MODE == 4)
        if (acc == 0x7fffffffffffLL) *sink = acc;  // keep acc alive
    if (MODE == 3) {
        __syncthreads();
        for (int i = threadIdx.x; i < 4096; i += blockDim.x)
            if (lh[i]) atomicAdd(&counts[i], lh[i]);
    }
}

// SoA variant of the read, for comparison with the AoS stride-24 loads
__global__ void kprobe_soa(const double* __restrict__ px,
                           const double* __restrict__ py,
                           const double* __restrict__ pz, int64_t n,
                           double inv, int64_t nm,
                           int* __restrict__ counts)
{
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < n; i += stride) {
        const int64_t c = cell_of(px[i], py[i], pz[i], inv, nm);
        atomicAdd(&counts[c], 1);
    }
}

int main(int argc, char** argv) {
    const int64_t n = argc > 1 ? atoll(argv[1]) : (int64_t)1e9;
    const int64_t nm = 1024;
    const int64_t ncells = nm * nm * nm;
    double *pos, *out;
    int* counts;
    int64_t* sink;
    CHECK(hipMalloc(&pos, 3 * n * sizeof(double)));
    CHECK(hipMalloc(&out, n * sizeof(double)));
    CHECK(hipMalloc(&counts, ncells * sizeof(int)));
    CHECK(hipMalloc(&sink, 8));

    // fill positions with a cheap scrambled pattern on device
    {
        std::vector<double> h(3 << 20);
        srand(1);
        for (auto& v : h) v = (double)rand() / RAND_MAX * 5000.0;
        for (int64_t off = 0; off < 3 * n; off += h.size()) {
            int64_t m = std::min((int64_t)h.size(), 3 * n - off);
            CHECK(hipMemcpy(pos + off, h.data(), m * 8,
                            hipMemcpyHostToDevice));
        }
    }
    const double inv = nm / 5000.0;
    const int grid = 8192, block = 256;

    auto bench = [&](const char* name, auto fn) {
        fn();  // warm
        CHECK(hipDeviceSynchronize());
        hipEvent_t a, b;
        hipEventCreate(&a); hipEventCreate(&b);
        hipEventRecord(a);
        for (int r = 0; r < 3; r++) fn();
        hipEventRecord(b);
        CHECK(hipEventSynchronize(b));
        float ms;
        hipEventElapsedTime(&ms, a, b);
        printf("%-8s %8.2f ms   (%.0f GB/s read)\n", name, ms / 3,
               24.0 * n / (ms / 3 / 1e3) / 1e9);
    };

    bench("read", [&] { hipLaunchKernelGGL(kprobe<0>, grid, block, 0, 0,
        pos, n, inv, nm, counts, sink, out); });
    bench("atom", [&] { hipLaunchKernelGGL(kprobe<1>, grid, block, 0, 0,
        pos, n, inv, nm, counts, sink, out); });
    bench("atomsm", [&] { hipLaunchKernelGGL(kprobe<2>, grid, block, 0, 0,
        pos, n, inv, nm, counts, sink, out); });
    bench("lds", [&] { hipLaunchKernelGGL(kprobe<3>, grid, block, 0, 0,
        pos, n, inv, nm, counts, sink, out); });
    bench("copy", [&] { hipLaunchKernelGGL(kprobe<4>, grid, block, 0, 0,
        pos, n, inv, nm, counts, sink, out); });
    bench("soa", [&] { hipLaunchKernelGGL(kprobe_soa, grid, block, 0, 0,
        pos, pos + n, pos + 2 * n, n, inv, nm, counts); });
    return 0;
}
