// Window-deposit ("paint") scatter kernels — the #1 kernel of the path
// (replaces pmesh's Cython scatter called at
// nbodykit/source/mesh/catalog.py:287,295-296).
//
// HBM-bound: algorithmic traffic 24 B position read + support^3 f64
// read-modify-writes per particle (CIC 152 B, TSC 456 B, PCS 1048 B).
// One thread per particle, grid-stride; SoA position reads are fully
// coalesced; deposits use hardware f64 global atomics (compile with
// -munsafe-fp-atomics so hipcc emits global_atomic_add_f64 instead of a
// CAS loop).  Window shapes are the B-splines fixed in-tree by their
// Fourier duals (source/mesh/catalog.py:453-594; Jing 2005 eq. 18 with
// p = 2/3/4); alignment: a particle exactly on a grid point deposits its
// full mass there.
#include "nbk_common.h"

namespace {

// Each thread owns a contiguous RUN of particles.  Catalogs on this path
// are usually cell-ordered (the LogNormal generator emits particles in
// global cell order, mirroring the reference's mpsort-by-cell-id,
// mockmaker.py:338-345): with one-thread-per-particle the 64 lanes of a
// wave then hit overlapping 8/27/64-cell neighbourhoods and the f64
// atomics serialize on shared addresses.  A run per thread spaces the
// lanes RUN cells apart, making intra-wave conflicts rare, while the
// deposits of one run stay L2-local.  Random-order catalogs are
// unaffected either way.
#define NBK_PAINT_RUN 8

template <int WINDOW>
__global__ void kpaint(const double* __restrict__ px,
                       const double* __restrict__ py,
                       const double* __restrict__ pz,
                       const double* __restrict__ mass, int64_t n,
                       int64_t n0, int64_t n1, int64_t n2,
                       double invH0, double invH1, double invH2,
                       double shift,
                       double* __restrict__ mesh,
                       int64_t x0, int64_t nx_local)
{
    const int64_t nruns = (n + NBK_PAINT_RUN - 1) / NBK_PAINT_RUN;
    const int64_t rstride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t run = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         run < nruns; run += rstride) {
        const int64_t ibeg = run * NBK_PAINT_RUN;
        const int64_t iend = min(ibeg + NBK_PAINT_RUN, n);
        for (int64_t i = ibeg; i < iend; i++) {
        const double u0 = px[i] * invH0 + shift;
        const double u1 = py[i] * invH1 + shift;
        const double u2 = pz[i] * invH2 + shift;
        const double m = mass ? mass[i] : 1.0;

        // per-axis cell offsets and weights
        constexpr int SUP = (WINDOW == NBK_WINDOW_CIC) ? 2
                          : (WINDOW == NBK_WINDOW_TSC) ? 3 : 4;
        double w0[SUP], w1[SUP], w2[SUP];
        int64_t b0, b1, b2;   // base cell per axis

        if (WINDOW == NBK_WINDOW_CIC) {
            const double f0 = floor(u0), f1 = floor(u1), f2 = floor(u2);
            b0 = (int64_t)f0; b1 = (int64_t)f1; b2 = (int64_t)f2;
            w0[1] = u0 - f0; w0[0] = 1.0 - w0[1];
            w1[1] = u1 - f1; w1[0] = 1.0 - w1[1];
            w2[1] = u2 - f2; w2[0] = 1.0 - w2[1];
        } else if (WINDOW == NBK_WINDOW_TSC) {
            // centered on the nearest grid point; support 3
            const double f0 = floor(u0 + 0.5), f1 = floor(u1 + 0.5),
                         f2 = floor(u2 + 0.5);
            b0 = (int64_t)f0 - 1; b1 = (int64_t)f1 - 1; b2 = (int64_t)f2 - 1;
            #pragma unroll
            for (int d = 0; d < 3; d++) {
                const double s0 = u0 - (f0 + d - 1);
                const double s1 = u1 - (f1 + d - 1);
                const double s2 = u2 - (f2 + d - 1);
                const double a0 = fabs(s0), a1 = fabs(s1), a2 = fabs(s2);
                w0[d] = a0 < 0.5 ? 0.75 - s0 * s0 : 0.5 * (1.5 - a0) * (1.5 - a0);
                w1[d] = a1 < 0.5 ? 0.75 - s1 * s1 : 0.5 * (1.5 - a1) * (1.5 - a1);
                w2[d] = a2 < 0.5 ? 0.75 - s2 * s2 : 0.5 * (1.5 - a2) * (1.5 - a2);
            }
        } else {  // PCS, support 4 (cubic B-spline)
            const double f0 = floor(u0), f1 = floor(u1), f2 = floor(u2);
            b0 = (int64_t)f0 - 1; b1 = (int64_t)f1 - 1; b2 = (int64_t)f2 - 1;
            #pragma unroll
            for (int d = 0; d < 4; d++) {
                const double s0 = fabs(u0 - (f0 + d - 1));
                const double s1 = fabs(u1 - (f1 + d - 1));
                const double s2 = fabs(u2 - (f2 + d - 1));
                w0[d] = s0 < 1.0 ? (4.0 - 6.0 * s0 * s0 + 3.0 * s0 * s0 * s0) / 6.0
                                 : (2.0 - s0) * (2.0 - s0) * (2.0 - s0) / 6.0;
                w1[d] = s1 < 1.0 ? (4.0 - 6.0 * s1 * s1 + 3.0 * s1 * s1 * s1) / 6.0
                                 : (2.0 - s1) * (2.0 - s1) * (2.0 - s1) / 6.0;
                w2[d] = s2 < 1.0 ? (4.0 - 6.0 * s2 * s2 + 3.0 * s2 * s2 * s2) / 6.0
                                 : (2.0 - s2) * (2.0 - s2) * (2.0 - s2) / 6.0;
            }
        }

        #pragma unroll
        for (int dx = 0; dx < SUP; dx++) {
            const int64_t gx = wrap_idx(b0 + dx, n0);
            if (gx < x0 || gx >= x0 + nx_local) continue;  // ghost-owned
            const int64_t lx = gx - x0;
            #pragma unroll
            for (int dy = 0; dy < SUP; dy++) {
                const int64_t gy = wrap_idx(b1 + dy, n1);
                const double wxy = w0[dx] * w1[dy] * m;
                #pragma unroll
                for (int dz = 0; dz < SUP; dz++) {
                    const int64_t gz = wrap_idx(b2 + dz, n2);
                    atomicAdd(&mesh[(lx * n1 + gy) * n2 + gz],
                              wxy * w2[dz]);
                }
            }
        }
        }
    }
}

__global__ void kreadout_nnb(const double* __restrict__ px,
                             const double* __restrict__ py,
                             const double* __restrict__ pz, int64_t n,
                             int64_t n0, int64_t n1, int64_t n2,
                             double invH0, double invH1, double invH2,
                             const double* __restrict__ mesh,
                             int64_t x0, int64_t nx_local,
                             double* __restrict__ out)
{
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < n; i += stride) {
        const int64_t gx = wrap_idx((int64_t)floor(px[i] * invH0 + 0.5), n0);
        const int64_t gy = wrap_idx((int64_t)floor(py[i] * invH1 + 0.5), n1);
        const int64_t gz = wrap_idx((int64_t)floor(pz[i] * invH2 + 0.5), n2);
        if (gx < x0 || gx >= x0 + nx_local) { out[i] = 0.0; continue; }
        out[i] = mesh[((gx - x0) * n1 + gy) * n2 + gz];
    }
}

int grid_for(int64_t n, int block) {
    int64_t g = (n + block - 1) / block;
    // >> 256 CUs x 8 XCDs want >>2048 workgroups; cap to keep index math sane
    if (g > 1048576) g = 1048576;
    if (g < 1) g = 1;
    return (int)g;
}

}  // namespace

extern "C" int nbk_paint_f64(const double* pos, const double* mass, int64_t n,
                             const int64_t nmesh[3], const double box[3],
                             int window, double shift,
                             double* mesh, int64_t x0, int64_t nx_local,
                             void* stream)
{
    if (n < 0 || !mesh || (!pos && n > 0)) {
        NBK_SET_ERR("nbk_paint_f64: bad pointer/size");
        return NBK_ERR_ARG;
    }
    if (n == 0) return NBK_OK;
    const double invH0 = nmesh[0] / box[0];
    const double invH1 = nmesh[1] / box[1];
    const double invH2 = nmesh[2] / box[2];
    const int block = 256;
    const int grid = grid_for((n + NBK_PAINT_RUN - 1) / NBK_PAINT_RUN,
                              block);
    hipStream_t s = (hipStream_t)stream;
    const double *px = pos, *py = pos + n, *pz = pos + 2 * n;

    switch (window) {
    case NBK_WINDOW_CIC:
        hipLaunchKernelGGL(kpaint<NBK_WINDOW_CIC>, dim3(grid), dim3(block), 0, s,
                           px, py, pz, mass, n, nmesh[0], nmesh[1], nmesh[2],
                           invH0, invH1, invH2, shift, mesh, x0, nx_local);
        break;
    case NBK_WINDOW_TSC:
        hipLaunchKernelGGL(kpaint<NBK_WINDOW_TSC>, dim3(grid), dim3(block), 0, s,
                           px, py, pz, mass, n, nmesh[0], nmesh[1], nmesh[2],
                           invH0, invH1, invH2, shift, mesh, x0, nx_local);
        break;
    case NBK_WINDOW_PCS:
        hipLaunchKernelGGL(kpaint<NBK_WINDOW_PCS>, dim3(grid), dim3(block), 0, s,
                           px, py, pz, mass, n, nmesh[0], nmesh[1], nmesh[2],
                           invH0, invH1, invH2, shift, mesh, x0, nx_local);
        break;
    default:
        NBK_SET_ERR("nbk_paint_f64: unknown window id %d", window);
        return NBK_ERR_ARG;
    }
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_readout_nnb_f64(const double* pos, int64_t n,
                                   const int64_t nmesh[3], const double box[3],
                                   const double* mesh, int64_t x0,
                                   int64_t nx_local,
                                   double* out, void* stream)
{
    if (n == 0) return NBK_OK;
    const int block = 256;
    const int grid = grid_for(n, block);
    hipStream_t s = (hipStream_t)stream;
    hipLaunchKernelGGL(kreadout_nnb, dim3(grid), dim3(block), 0, s,
                       pos, pos + n, pos + 2 * n, n,
                       nmesh[0], nmesh[1], nmesh[2],
                       nmesh[0] / box[0], nmesh[1] / box[1], nmesh[2] / box[2],
                       mesh, x0, nx_local, out);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}
