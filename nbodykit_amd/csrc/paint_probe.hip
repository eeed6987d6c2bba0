// Standalone paint-kernel ablation probe (not part of libnbk_hip).
// Times variants of the CIC deposit on synthetic cell-ordered particles
// (the C4 shape: ~1 particle/cell, 1024^3 f64 mesh) to find the
// bottleneck: atomic serialization vs atomic throughput vs read path.
//
//   hipcc --offload-arch=gfx950 -O3 -munsafe-fp-atomics paint_probe.hip -o paint_probe
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

#define CHECK(x) do { hipError_t e=(x); if(e){printf("ERR %s %s\n",#x,hipGetErrorString(e)); exit(1);} } while(0)

// mode 0: sorted uniform (1/cell); 1: sorted clumpy (20% of particles
// in ~1000-deep single-cell clumps); 2: random shuffled
__global__ void gen_sorted(double* px, double* py, double* pz, long n,
                           long N, double H, unsigned seed, int mode)
{
    long i = blockIdx.x * (long)blockDim.x + threadIdx.x;
    const long stride = (long)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        // cell = i * ncells / n in C order, plus a cheap hash jitter
        const double ratio = (double)N * N * N / (double)n;
        long cell = (long)((double)i * ratio);
        if (mode == 1) {
            unsigned hc = (unsigned)(((unsigned long)(i >> 10)) * 2246822519u);
            if ((hc % 5u) == 0u)  // whole 1024-particle block -> one cell
                cell = (long)((double)((i >> 10) << 10) * ratio);
        } else if (mode == 2) {
            unsigned h1 = (unsigned)(i * 2654435761L) ^ 0x9e3779b9u;
            long c2 = ((long)h1 << 18) ^ (i * 1103515245L);
            cell = c2 % ((long)N * N * N);
            if (cell < 0) cell += (long)N * N * N;
        }
        long iz = cell % N, iy = (cell / N) % N, ix = cell / (N * N);
        unsigned h = (unsigned)(i * 2654435761u) ^ seed;
        double f1 = (h & 1023) / 1024.0;
        double f2 = ((h >> 10) & 1023) / 1024.0;
        double f3 = ((h >> 20) & 1023) / 1024.0;
        px[i] = (ix + f1) * H;
        py[i] = (iy + f2) * H;
        pz[i] = (iz + f3) * H;
    }
}

__device__ __forceinline__ long wrapi(long i, long n) {
    i %= n;
    return i < 0 ? i + n : i;
}

// variant A: one thread per particle, 8 global f64 atomics
template <int RUN, bool ATOMIC>
__global__ void paint_cic(const double* __restrict__ px,
                          const double* __restrict__ py,
                          const double* __restrict__ pz, long n,
                          long N, double invH, double* __restrict__ mesh)
{
    const long nruns = (n + RUN - 1) / RUN;
    const long stride = (long)gridDim.x * blockDim.x;
    for (long r = blockIdx.x * (long)blockDim.x + threadIdx.x;
         r < nruns; r += stride) {
        const long ibeg = (long)r * RUN;
        const long iend = min(ibeg + RUN, n);
        for (long i = ibeg; i < iend; i++) {
            const double u0 = px[i] * invH, u1 = py[i] * invH,
                         u2 = pz[i] * invH;
            const double f0 = floor(u0), f1 = floor(u1), f2 = floor(u2);
            const double a0 = u0 - f0, a1 = u1 - f1, a2 = u2 - f2;
            const long b0 = (long)f0, b1 = (long)f1, b2 = (long)f2;
            const double w0[2] = {1.0 - a0, a0};
            const double w1[2] = {1.0 - a1, a1};
            const double w2[2] = {1.0 - a2, a2};
            for (int dx = 0; dx < 2; dx++) {
                const long gx = wrapi(b0 + dx, N);
                for (int dy = 0; dy < 2; dy++) {
                    const long gy = wrapi(b1 + dy, N);
                    const double wxy = w0[dx] * w1[dy];
                    for (int dz = 0; dz < 2; dz++) {
                        const long gz = wrapi(b2 + dz, N);
                        double* p = &mesh[(gx * N + gy) * N + gz];
                        if (ATOMIC) atomicAdd(p, wxy * w2[dz]);
                        else *p += wxy * w2[dz];
                    }
                }
            }
        }
    }
}

// variant: stride mapping + wave-segmented merge of equal deposit
// addresses (sorted input puts same-cell particles in adjacent lanes)
__global__ void paint_cic_merge(const double* __restrict__ px,
                                const double* __restrict__ py,
                                const double* __restrict__ pz, long n,
                                long N, double invH,
                                double* __restrict__ mesh)
{
    const long stride = (long)gridDim.x * blockDim.x;
    const int lane = threadIdx.x & 63;
    for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
         i += stride) {
        const double u0 = px[i] * invH, u1 = py[i] * invH,
                     u2 = pz[i] * invH;
        const double f0 = floor(u0), f1 = floor(u1), f2 = floor(u2);
        const double a0 = u0 - f0, a1 = u1 - f1, a2 = u2 - f2;
        const long b0 = (long)f0, b1 = (long)f1, b2 = (long)f2;
        const double w0[2] = {1.0 - a0, a0};
        const double w1[2] = {1.0 - a1, a1};
        const double w2[2] = {1.0 - a2, a2};
        for (int dx = 0; dx < 2; dx++) {
            const long gx = wrapi(b0 + dx, N);
            for (int dy = 0; dy < 2; dy++) {
                const long gy = wrapi(b1 + dy, N);
                const double wxy = w0[dx] * w1[dy];
                for (int dz = 0; dz < 2; dz++) {
                    const long gz = wrapi(b2 + dz, N);
                    long addr = (gx * N + gy) * N + gz;
                    double val = wxy * w2[dz];
                    // segmented inclusive prefix over equal-addr runs
                    #pragma unroll
                    for (int d = 1; d < 64; d <<= 1) {
                        long a_up = __shfl_up((long long)addr, d, 64);
                        double v_up = __shfl_up(val, d, 64);
                        if (lane >= d && a_up == addr) val += v_up;
                    }
                    long a_dn = __shfl_down((long long)addr, 1, 64);
                    if (lane == 63 || a_dn != addr)
                        atomicAdd(&mesh[addr], val);
                }
            }
        }
    }
}

// TSC with/without the wave merge (C3 shape diagnosis)
__device__ __forceinline__ void probe_merged_deposit(double* mesh,
                                                     long long addr,
                                                     double val, int lane) {
    const long long a_up1 = __shfl_up(addr, 1, 64);
    const bool head = (lane == 0) || (a_up1 != addr);
    const unsigned long long heads = __ballot(head);
    const unsigned long long below = heads & (~0ULL >> (63 - lane));
    const int myhead = 63 - __clzll(below);
    #pragma unroll
    for (int d = 1; d < 64; d <<= 1) {
        const double v_up = __shfl_up(val, d, 64);
        if (lane - d >= myhead) val += v_up;
    }
    const bool next_head = ((lane < 63) && ((heads >> (lane + 1)) & 1ULL));
    if (addr >= 0 && (lane == 63 || next_head)) atomicAdd(&mesh[addr], val);
}

template <bool MERGE>
__global__ void paint_tsc(const double* __restrict__ px,
                          const double* __restrict__ py,
                          const double* __restrict__ pz, long n,
                          long N, double invH, double* __restrict__ mesh)
{
    const int lane = threadIdx.x & 63;
    const long stride = (long)gridDim.x * blockDim.x;
    const long wbase0 = blockIdx.x * (long)blockDim.x + (threadIdx.x & ~63);
    for (long wb = wbase0; wb < n; wb += stride) {
        const long i = wb + lane;
        const bool valid = i < n;
        const double u0 = valid ? px[i] * invH : 0.0;
        const double u1 = valid ? py[i] * invH : 0.0;
        const double u2 = valid ? pz[i] * invH : 0.0;
        const double f0 = floor(u0 + 0.5), f1 = floor(u1 + 0.5),
                     f2 = floor(u2 + 0.5);
        const long b0 = (long)f0 - 1, b1 = (long)f1 - 1, b2 = (long)f2 - 1;
        double w0[3], w1[3], w2[3];
        #pragma unroll
        for (int d = 0; d < 3; d++) {
            const double s0 = u0 - (f0 + d - 1), s1 = u1 - (f1 + d - 1),
                         s2 = u2 - (f2 + d - 1);
            const double a0 = fabs(s0), a1 = fabs(s1), a2 = fabs(s2);
            w0[d] = a0 < 0.5 ? 0.75 - s0*s0 : 0.5*(1.5-a0)*(1.5-a0);
            w1[d] = a1 < 0.5 ? 0.75 - s1*s1 : 0.5*(1.5-a1)*(1.5-a1);
            w2[d] = a2 < 0.5 ? 0.75 - s2*s2 : 0.5*(1.5-a2)*(1.5-a2);
        }
        #pragma unroll
        for (int dx = 0; dx < 3; dx++) {
            const long gx = wrapi(b0 + dx, N);
            #pragma unroll
            for (int dy = 0; dy < 3; dy++) {
                const long gy = wrapi(b1 + dy, N);
                const double wxy = w0[dx] * w1[dy];
                #pragma unroll
                for (int dz = 0; dz < 3; dz++) {
                    const long gz = wrapi(b2 + dz, N);
                    if (MERGE) {
                        const long long addr = valid
                            ? (long long)((gx * N + gy) * N + gz)
                            : (long long)(-1 - lane);
                        probe_merged_deposit(mesh, addr, wxy * w2[dz], lane);
                    } else if (valid) {
                        atomicAdd(&mesh[(gx * N + gy) * N + gz],
                                  wxy * w2[dz]);
                    }
                }
            }
        }
    }
}

// hoisted-address TSC: per-axis wraps computed once, 32-bit row bases,
// add-only inner loop — attacks the 71-88% instruction-issue stall
template <bool MERGE>
__global__ void paint_tsc_hoisted(const double* __restrict__ px,
                                  const double* __restrict__ py,
                                  const double* __restrict__ pz, long n,
                                  long N, double invH,
                                  double* __restrict__ mesh)
{
    const int lane = threadIdx.x & 63;
    const long stride = (long)gridDim.x * blockDim.x;
    const long wbase0 = blockIdx.x * (long)blockDim.x + (threadIdx.x & ~63);
    const unsigned uN = (unsigned)N;
    for (long wb = wbase0; wb < n; wb += stride) {
        const long i = wb + lane;
        const bool valid = i < n;
        const double u0 = valid ? px[i]*invH : 0.0;
        const double u1 = valid ? py[i]*invH : 0.0;
        const double u2 = valid ? pz[i]*invH : 0.0;
        const double f0 = floor(u0+0.5), f1 = floor(u1+0.5), f2 = floor(u2+0.5);
        double w0[3], w1[3], w2[3];
        #pragma unroll
        for (int d=0; d<3; d++) {
            double s0=u0-(f0+d-1), s1=u1-(f1+d-1), s2=u2-(f2+d-1);
            double a0=fabs(s0),a1=fabs(s1),a2=fabs(s2);
            w0[d]=a0<0.5?0.75-s0*s0:0.5*(1.5-a0)*(1.5-a0);
            w1[d]=a1<0.5?0.75-s1*s1:0.5*(1.5-a1)*(1.5-a1);
            w2[d]=a2<0.5?0.75-s2*s2:0.5*(1.5-a2)*(1.5-a2);
        }
        unsigned gx[3], gy[3], gz[3];
        #pragma unroll
        for (int d=0; d<3; d++) {
            int v0 = (int)f0 - 1 + d; v0 = v0 < 0 ? v0 + (int)N : (v0 >= N ? v0 - (int)N : v0);
            int v1 = (int)f1 - 1 + d; v1 = v1 < 0 ? v1 + (int)N : (v1 >= N ? v1 - (int)N : v1);
            int v2 = (int)f2 - 1 + d; v2 = v2 < 0 ? v2 + (int)N : (v2 >= N ? v2 - (int)N : v2);
            gx[d] = (unsigned)v0; gy[d] = (unsigned)v1; gz[d] = (unsigned)v2;
        }
        #pragma unroll
        for (int dx=0; dx<3; dx++) {
            const unsigned xbase = gx[dx] * uN;
            #pragma unroll
            for (int dy=0; dy<3; dy++) {
                const unsigned rb = (xbase + gy[dy]) * uN;
                const double wxy = w0[dx]*w1[dy];
                #pragma unroll
                for (int dz=0; dz<3; dz++) {
                    const double val = wxy*w2[dz];
                    if (MERGE) {
                        long long addr = valid ? (long long)(rb + gz[dz])
                                               : (long long)(-1 - lane);
                        // head-bounded segmented merge
                        const long long a_up1 = __shfl_up(addr, 1, 64);
                        const bool head = (lane == 0) || (a_up1 != addr);
                        const unsigned long long heads = __ballot(head);
                        const unsigned long long below = heads & (~0ULL >> (63 - lane));
                        const int myhead = 63 - __clzll(below);
                        double v = val;
                        #pragma unroll
                        for (int d = 1; d < 64; d <<= 1) {
                            const double v_up = __shfl_up(v, d, 64);
                            if (lane - d >= myhead) v += v_up;
                        }
                        const bool nh = (lane < 63) && ((heads >> (lane+1)) & 1ULL);
                        if (addr >= 0 && (lane == 63 || nh))
                            atomicAdd(&mesh[addr], v);
                    } else if (valid) {
                        atomicAdd(&mesh[rb + gz[dz]], val);
                    }
                }
            }
        }
    }
}

// tiled TSC (LDS window per sorted block) — mirror of kpaint_tiled
#define PPB 1024
#define MAXC 8192
__global__ void paint_tsc_tiled(const double* __restrict__ px,
                                const double* __restrict__ py,
                                const double* __restrict__ pz, long n,
                                long N, double invH,
                                double* __restrict__ mesh)
{
    __shared__ double buf[MAXC];
    __shared__ int s_min0, s_max0, s_min1, s_max1;
    const long ibeg = (long)blockIdx.x * PPB;
    const long iend = min(ibeg + PPB, n);
    if (threadIdx.x == 0) {
        s_min0 = INT_MAX; s_max0 = INT_MIN;
        s_min1 = INT_MAX; s_max1 = INT_MIN;
    }
    __syncthreads();
    int mn0 = INT_MAX, mx0 = INT_MIN, mn1 = INT_MAX, mx1 = INT_MIN;
    for (long i = ibeg + threadIdx.x; i < iend; i += blockDim.x) {
        const int b0 = (int)floor(px[i] * invH + 0.5) - 1;
        const int b1 = (int)floor(py[i] * invH + 0.5) - 1;
        mn0 = min(mn0, b0); mx0 = max(mx0, b0);
        mn1 = min(mn1, b1); mx1 = max(mx1, b1);
    }
    atomicMin(&s_min0, mn0); atomicMax(&s_max0, mx0);
    atomicMin(&s_min1, mn1); atomicMax(&s_max1, mx1);
    __syncthreads();
    const int wx = s_max0 - s_min0 + 3;
    const int wy = s_max1 - s_min1 + 3;
    const long cells = (long)wx * wy * N;
    if (ibeg >= iend) return;
    if (cells <= 0 || cells > MAXC) {
        for (long i = ibeg + threadIdx.x; i < iend; i += blockDim.x) {
            const double u0 = px[i]*invH, u1 = py[i]*invH, u2 = pz[i]*invH;
            const double f0 = floor(u0+0.5), f1 = floor(u1+0.5), f2 = floor(u2+0.5);
            const long b0=(long)f0-1, b1=(long)f1-1, b2=(long)f2-1;
            double w0[3], w1[3], w2[3];
            for (int d=0; d<3; d++) {
                double s0=u0-(f0+d-1), s1=u1-(f1+d-1), s2=u2-(f2+d-1);
                double a0=fabs(s0),a1=fabs(s1),a2=fabs(s2);
                w0[d]=a0<0.5?0.75-s0*s0:0.5*(1.5-a0)*(1.5-a0);
                w1[d]=a1<0.5?0.75-s1*s1:0.5*(1.5-a1)*(1.5-a1);
                w2[d]=a2<0.5?0.75-s2*s2:0.5*(1.5-a2)*(1.5-a2);
            }
            for (int dx=0;dx<3;dx++){long gx=wrapi(b0+dx,N);
              for(int dy=0;dy<3;dy++){long gy=wrapi(b1+dy,N);
                double wxy=w0[dx]*w1[dy];
                for(int dz=0;dz<3;dz++){long gz=wrapi(b2+dz,N);
                  atomicAdd(&mesh[(gx*N+gy)*N+gz], wxy*w2[dz]);}}}
        }
        return;
    }
    for (long w = threadIdx.x; w < cells; w += blockDim.x) buf[w] = 0.0;
    __syncthreads();
    for (long i = ibeg + threadIdx.x; i < iend; i += blockDim.x) {
        const double u0 = px[i]*invH, u1 = py[i]*invH, u2 = pz[i]*invH;
        const double f0 = floor(u0+0.5), f1 = floor(u1+0.5), f2 = floor(u2+0.5);
        const long b0=(long)f0-1, b1=(long)f1-1, b2=(long)f2-1;
        double w0[3], w1[3], w2[3];
        for (int d=0; d<3; d++) {
            double s0=u0-(f0+d-1), s1=u1-(f1+d-1), s2=u2-(f2+d-1);
            double a0=fabs(s0),a1=fabs(s1),a2=fabs(s2);
            w0[d]=a0<0.5?0.75-s0*s0:0.5*(1.5-a0)*(1.5-a0);
            w1[d]=a1<0.5?0.75-s1*s1:0.5*(1.5-a1)*(1.5-a1);
            w2[d]=a2<0.5?0.75-s2*s2:0.5*(1.5-a2)*(1.5-a2);
        }
        const int lx = (int)(b0 - s_min0);
        const int ly = (int)(b1 - s_min1);
        for (int dx=0;dx<3;dx++)
          for (int dy=0;dy<3;dy++){
            const double wxy=w0[dx]*w1[dy];
            const long base=((long)(lx+dx)*wy+(ly+dy))*N;
            for (int dz=0;dz<3;dz++){
              const long gz=wrapi(b2+dz,N);
              unsafeAtomicAdd(&buf[base+gz], wxy*w2[dz]);
            }
          }
    }
    __syncthreads();
    for (long w = threadIdx.x; w < cells; w += blockDim.x) {
        const double v = buf[w];
        if (v == 0.0) continue;
        const long gz = w % N;
        const long wyi = (w / N) % wy;
        const long wxi = w / ((long)N * wy);
        const long gx = wrapi(s_min0 + wxi, N);
        const long gy = wrapi(s_min1 + wyi, N);
        atomicAdd(&mesh[(gx*N+gy)*N+gz], v);
    }
}

// variant: read-only (bandwidth leg)
__global__ void read_only(const double* __restrict__ px,
                          const double* __restrict__ py,
                          const double* __restrict__ pz, long n,
                          double* __restrict__ sink)
{
    double acc = 0;
    const long stride = (long)gridDim.x * blockDim.x;
    for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
         i += stride)
        acc += px[i] + py[i] + pz[i];
    if (acc == 12345.6789) *sink = acc;   // never true; defeats DCE
}

// variant: z-column register accumulation — each thread walks RUN
// consecutive (sorted) particles keeping a [2][2] x (z,z+1) window in
// registers, flushing atomics only when the base cell moves.
template <int RUN>
__global__ void paint_cic_window(const double* __restrict__ px,
                                 const double* __restrict__ py,
                                 const double* __restrict__ pz, long n,
                                 long N, double invH,
                                 double* __restrict__ mesh)
{
    const long nruns = (n + RUN - 1) / RUN;
    const long stride = (long)gridDim.x * blockDim.x;
    for (long r = blockIdx.x * (long)blockDim.x + threadIdx.x;
         r < nruns; r += stride) {
        const long ibeg = (long)r * RUN;
        const long iend = min(ibeg + RUN, n);
        long cb0 = -1, cb1 = -1, cb2 = -1;     // current window base
        double acc[2][2][2] = {};
        for (long i = ibeg; i < iend; i++) {
            const double u0 = px[i] * invH, u1 = py[i] * invH,
                         u2 = pz[i] * invH;
            const double f0 = floor(u0), f1 = floor(u1), f2 = floor(u2);
            const double a0 = u0 - f0, a1 = u1 - f1, a2 = u2 - f2;
            const long b0 = (long)f0, b1 = (long)f1, b2 = (long)f2;
            if (b0 != cb0 || b1 != cb1 || b2 != cb2) {
                if (cb0 >= 0) {
                    for (int dx = 0; dx < 2; dx++)
                        for (int dy = 0; dy < 2; dy++)
                            for (int dz = 0; dz < 2; dz++)
                                if (acc[dx][dy][dz] != 0.0)
                                    atomicAdd(&mesh[(wrapi(cb0+dx,N)*N
                                        + wrapi(cb1+dy,N))*N
                                        + wrapi(cb2+dz,N)],
                                        acc[dx][dy][dz]);
                }
                for (int dx = 0; dx < 2; dx++)
                    for (int dy = 0; dy < 2; dy++)
                        for (int dz = 0; dz < 2; dz++)
                            acc[dx][dy][dz] = 0.0;
                cb0 = b0; cb1 = b1; cb2 = b2;
            }
            const double w0[2] = {1.0 - a0, a0};
            const double w1[2] = {1.0 - a1, a1};
            const double w2[2] = {1.0 - a2, a2};
            for (int dx = 0; dx < 2; dx++)
                for (int dy = 0; dy < 2; dy++)
                    for (int dz = 0; dz < 2; dz++)
                        acc[dx][dy][dz] += w0[dx] * w1[dy] * w2[dz];
        }
        if (cb0 >= 0)
            for (int dx = 0; dx < 2; dx++)
                for (int dy = 0; dy < 2; dy++)
                    for (int dz = 0; dz < 2; dz++)
                        if (acc[dx][dy][dz] != 0.0)
                            atomicAdd(&mesh[(wrapi(cb0+dx,N)*N
                                + wrapi(cb1+dy,N))*N + wrapi(cb2+dz,N)],
                                acc[dx][dy][dz]);
    }
}

int grid_for(long work) {
    long g = (work + 255) / 256;
    if (g > 1048576) g = 1048576;
    return (int)(g < 1 ? 1 : g);
}

template <typename F>
double timeit(F&& launch, int iters) {
    hipEvent_t a, b;
    CHECK(hipEventCreate(&a));
    CHECK(hipEventCreate(&b));
    launch();                      // warm
    CHECK(hipDeviceSynchronize());
    CHECK(hipEventRecord(a));
    for (int it = 0; it < iters; it++) launch();
    CHECK(hipEventRecord(b));
    CHECK(hipEventSynchronize(b));
    float ms;
    CHECK(hipEventElapsedTime(&ms, a, b));
    return ms / iters;
}

int main(int argc, char** argv) {
    const long N = argc > 1 ? atol(argv[1]) : 1024;
    const long n = argc > 2 ? atol(argv[2]) : 1000000000L;
    const double L = 5000.0, H = L / N, invH = N / L;

    double *px, *py, *pz, *mesh, *sink;
    CHECK(hipMalloc(&px, n * 8));
    CHECK(hipMalloc(&py, n * 8));
    CHECK(hipMalloc(&pz, n * 8));
    CHECK(hipMalloc(&mesh, N * N * N * 8));
    CHECK(hipMalloc(&sink, 8));

    const double GB = n * 152.0 / 1e9;   // algorithmic CIC bytes
    auto report = [&](const char* name, double ms) {
        printf("%-28s %8.2f ms  %7.1f GB/s(algo)  %6.2f Gpart/s\n",
               name, ms, GB / ms * 1000.0 / 1.0, n / ms / 1e6 / 1000.0);
    };

    const char* modenames[3] = {"sorted-uniform", "sorted-clumpy", "random"};
    // TSC A/B at the C3 shape (pass N=512 n=1e8 on the command line)
    for (int mode = 0; mode < 2; mode++) {
        hipLaunchKernelGGL(gen_sorted, dim3(grid_for(n)), dim3(256), 0, 0,
                           px, py, pz, n, N, H, 999u, mode);
        CHECK(hipDeviceSynchronize());
        char buf[128];
        snprintf(buf, 128, "tsc merged [%s]", mode ? "clumpy" : "uniform");
        report(buf, timeit([&] {
            hipLaunchKernelGGL((paint_tsc<true>), dim3(grid_for(n)),
                               dim3(256), 0, 0, px, py, pz, n, N, invH, mesh);
        }, 3));
        snprintf(buf, 128, "tsc plain  [%s]", mode ? "clumpy" : "uniform");
        report(buf, timeit([&] {
            hipLaunchKernelGGL((paint_tsc<false>), dim3(grid_for(n)),
                               dim3(256), 0, 0, px, py, pz, n, N, invH, mesh);
        }, 3));
        snprintf(buf, 128, "tsc hoist-plain [%s]", mode ? "clumpy" : "uniform");
        report(buf, timeit([&] {
            hipLaunchKernelGGL((paint_tsc_hoisted<false>), dim3(grid_for(n)),
                               dim3(256), 0, 0, px, py, pz, n, N, invH, mesh);
        }, 3));
        snprintf(buf, 128, "tsc hoist-merge [%s]", mode ? "clumpy" : "uniform");
        report(buf, timeit([&] {
            hipLaunchKernelGGL((paint_tsc_hoisted<true>), dim3(grid_for(n)),
                               dim3(256), 0, 0, px, py, pz, n, N, invH, mesh);
        }, 3));
        snprintf(buf, 128, "tsc tiled  [%s]", mode ? "clumpy" : "uniform");
        report(buf, timeit([&] {
            hipLaunchKernelGGL(paint_tsc_tiled,
                               dim3((unsigned)((n + PPB - 1) / PPB)),
                               dim3(256), 0, 0, px, py, pz, n, N, invH, mesh);
        }, 3));
    }
    for (int mode = 0; mode < 3; mode++) {
        hipLaunchKernelGGL(gen_sorted, dim3(grid_for(n)), dim3(256), 0, 0,
                           px, py, pz, n, N, H, 12345u, mode);
        CHECK(hipDeviceSynchronize());
        char buf[128];
        snprintf(buf, 128, "atomic RUN=1 [%s]", modenames[mode]);
        report(buf, timeit([&] {
            hipLaunchKernelGGL((paint_cic<1, true>), dim3(grid_for(n)),
                               dim3(256), 0, 0, px, py, pz, n, N, invH, mesh);
        }, 3));
        snprintf(buf, 128, "wave-merge [%s]", modenames[mode]);
        report(buf, timeit([&] {
            hipLaunchKernelGGL(paint_cic_merge, dim3(grid_for(n)),
                               dim3(256), 0, 0, px, py, pz, n, N, invH, mesh);
        }, 3));
    }
    return 0;
}
