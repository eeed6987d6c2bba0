// Window-deposit ("paint") kernels — the #1 op of the path (replaces
// pmesh's Cython scatter called at
// nbodykit/source/mesh/catalog.py:287,295-296).  Two strategies:
//
// 1. kpaint_gather (+optional fused z-FFT): the BIG-MESH path, fed by
//    the atomic-free two-level locality sort (nbk_sort.hip).  Each
//    block owns an exclusive LDS mesh tile (RG y-rows x full z of one
//    x-plane), gathers the source rows whose stencils can reach it via
//    the sort's row table, deposits with LDS ds_add_f64 and flushes
//    with plain stores — ZERO global atomics (the global atomic pipe
//    measures ~25 G op/s regardless of locality, csrc/count_probe.hip,
//    which bounded the scatter kernel below at 47 ms for 1e9 CIC).
//    The DOFFT variant runs the forward z-FFT on the tile's rows
//    before flushing, so the real mesh never exists in HBM
//    (CatalogMesh.to_complex_field).
//
// 2. kpaint: the scatter fallback for small chunks/meshes.  One thread
//    per particle in wave-contiguous order; deposits use hardware f64
//    global atomics (-munsafe-fp-atomics => global_atomic_add_f64)
//    with a wave-level segmented merge: adjacent lanes holding the
//    same target address combine via a shfl prefix-sum and only the
//    run tail issues the atomic (probe: clumpy 167 -> 45 ms).
//
// Window shapes are the B-splines fixed in-tree by their Fourier duals
// (source/mesh/catalog.py:453-594; Jing 2005 eq. 18, p = 2/3/4);
// a particle exactly on a grid point deposits its full mass there.
#include "nbk_common.h"

namespace {

// one deposit with wave-segmented address merge; addr < 0 = inactive
// lane (tail of the grid or ghost-owned cell) — must still participate
// in the shuffles.
__device__ __forceinline__ void merged_deposit(double* __restrict__ mesh,
                                               long long addr, double val,
                                               int lane) {
    // segment = maximal CONTIGUOUS run of equal addresses.  A plain
    // "same addr at distance d" guard would also fold non-adjacent
    // duplicates ([A A B A] merging the stray A into the first run);
    // the head-bounded Hillis-Steele scan below only sums within the
    // lane's own run.
    const long long a_up1 = __shfl_up(addr, 1, 64);
    const bool head = (lane == 0) || (a_up1 != addr);
    const unsigned long long heads = __ballot(head);
    const unsigned long long below = heads
        & (~0ULL >> (63 - lane));          // head bits at lanes <= lane
    const int myhead = 63 - __clzll(below);
    #pragma unroll
    for (int d = 1; d < 64; d <<= 1) {
        const double v_up = __shfl_up(val, d, 64);
        if (lane - d >= myhead) val += v_up;
    }
    const bool next_head = ((lane < 63)
                            && ((heads >> (lane + 1)) & 1ULL));
    if (addr >= 0 && (lane == 63 || next_head))
        atomicAdd(&mesh[addr], val);
}

// per-axis window weights + unwrapped base cell (shared by the tiled
// kernel's two paths; the main kpaint keeps its inlined version)
template <int WINDOW, int SUP>
__device__ __forceinline__ void paint_weights(double u0, double u1,
                                              double u2,
                                              double (&w0)[SUP],
                                              double (&w1)[SUP],
                                              double (&w2)[SUP],
                                              int64_t& b0, int64_t& b1,
                                              int64_t& b2) {
    if (WINDOW == NBK_WINDOW_CIC) {
        const double f0 = floor(u0), f1 = floor(u1), f2 = floor(u2);
        b0 = (int64_t)f0; b1 = (int64_t)f1; b2 = (int64_t)f2;
        w0[SUP - 1] = u0 - f0; w0[0] = 1.0 - (u0 - f0);
        w1[SUP - 1] = u1 - f1; w1[0] = 1.0 - (u1 - f1);
        w2[SUP - 1] = u2 - f2; w2[0] = 1.0 - (u2 - f2);
    } else if (WINDOW == NBK_WINDOW_TSC) {
        const double f0 = floor(u0 + 0.5), f1 = floor(u1 + 0.5),
                     f2 = floor(u2 + 0.5);
        b0 = (int64_t)f0 - 1; b1 = (int64_t)f1 - 1; b2 = (int64_t)f2 - 1;
        #pragma unroll
        for (int d = 0; d < 3; d++) {
            const double s0 = u0 - (f0 + d - 1);
            const double s1 = u1 - (f1 + d - 1);
            const double s2 = u2 - (f2 + d - 1);
            const double a0 = fabs(s0), a1 = fabs(s1), a2 = fabs(s2);
            w0[d] = a0 < 0.5 ? 0.75 - s0 * s0
                             : 0.5 * (1.5 - a0) * (1.5 - a0);
            w1[d] = a1 < 0.5 ? 0.75 - s1 * s1
                             : 0.5 * (1.5 - a1) * (1.5 - a1);
            w2[d] = a2 < 0.5 ? 0.75 - s2 * s2
                             : 0.5 * (1.5 - a2) * (1.5 - a2);
        }
    } else {   // PCS
        const double f0 = floor(u0), f1 = floor(u1), f2 = floor(u2);
        b0 = (int64_t)f0 - 1; b1 = (int64_t)f1 - 1; b2 = (int64_t)f2 - 1;
        #pragma unroll
        for (int d = 0; d < 4; d++) {
            const double s0 = fabs(u0 - (f0 + d - 1));
            const double s1 = fabs(u1 - (f1 + d - 1));
            const double s2 = fabs(u2 - (f2 + d - 1));
            w0[d] = s0 < 1.0
                ? (4.0 - 6.0 * s0 * s0 + 3.0 * s0 * s0 * s0) / 6.0
                : (2.0 - s0) * (2.0 - s0) * (2.0 - s0) / 6.0;
            w1[d] = s1 < 1.0
                ? (4.0 - 6.0 * s1 * s1 + 3.0 * s1 * s1 * s1) / 6.0
                : (2.0 - s1) * (2.0 - s1) * (2.0 - s1) / 6.0;
            w2[d] = s2 < 1.0
                ? (4.0 - 6.0 * s2 * s2 + 3.0 * s2 * s2 * s2) / 6.0
                : (2.0 - s2) * (2.0 - s2) * (2.0 - s2) / 6.0;
        }
    }
}

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
    constexpr int SUP = (WINDOW == NBK_WINDOW_CIC) ? 2
                      : (WINDOW == NBK_WINDOW_TSC) ? 3 : 4;
    const int lane = threadIdx.x & 63;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    // iterate whole waves so the shuffle stays collective at the tail
    const int64_t wbase0 = blockIdx.x * (int64_t)blockDim.x
        + (threadIdx.x & ~63);
    for (int64_t wb = wbase0; wb < n; wb += stride) {
        const int64_t i = wb + lane;
        const bool valid = i < n;

        const double u0 = valid ? px[i] * invH0 + shift : 0.0;
        const double u1 = valid ? py[i] * invH1 + shift : 0.0;
        const double u2 = valid ? pz[i] * invH2 + shift : 0.0;
        const double m = valid ? (mass ? mass[i] : 1.0) : 0.0;

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
                w0[d] = a0 < 0.5 ? 0.75 - s0 * s0
                                 : 0.5 * (1.5 - a0) * (1.5 - a0);
                w1[d] = a1 < 0.5 ? 0.75 - s1 * s1
                                 : 0.5 * (1.5 - a1) * (1.5 - a1);
                w2[d] = a2 < 0.5 ? 0.75 - s2 * s2
                                 : 0.5 * (1.5 - a2) * (1.5 - a2);
            }
        } else {  // PCS, support 4 (cubic B-spline)
            const double f0 = floor(u0), f1 = floor(u1), f2 = floor(u2);
            b0 = (int64_t)f0 - 1; b1 = (int64_t)f1 - 1; b2 = (int64_t)f2 - 1;
            #pragma unroll
            for (int d = 0; d < 4; d++) {
                const double s0 = fabs(u0 - (f0 + d - 1));
                const double s1 = fabs(u1 - (f1 + d - 1));
                const double s2 = fabs(u2 - (f2 + d - 1));
                w0[d] = s0 < 1.0
                    ? (4.0 - 6.0 * s0 * s0 + 3.0 * s0 * s0 * s0) / 6.0
                    : (2.0 - s0) * (2.0 - s0) * (2.0 - s0) / 6.0;
                w1[d] = s1 < 1.0
                    ? (4.0 - 6.0 * s1 * s1 + 3.0 * s1 * s1 * s1) / 6.0
                    : (2.0 - s1) * (2.0 - s1) * (2.0 - s1) / 6.0;
                w2[d] = s2 < 1.0
                    ? (4.0 - 6.0 * s2 * s2 + 3.0 * s2 * s2 * s2) / 6.0
                    : (2.0 - s2) * (2.0 - s2) * (2.0 - s2) / 6.0;
            }
        }

        #pragma unroll
        for (int dx = 0; dx < SUP; dx++) {
            const int64_t gx = wrap_idx(b0 + dx, n0);
            // ghost-owned cells: keep the lane in the wave ops with a
            // unique negative sentinel so it never merges or deposits
            const bool in_slab = valid && gx >= x0 && gx < x0 + nx_local;
            const int64_t lx = gx - x0;
            #pragma unroll
            for (int dy = 0; dy < SUP; dy++) {
                const int64_t gy = wrap_idx(b1 + dy, n1);
                const double wxy = w0[dx] * w1[dy] * m;
                #pragma unroll
                for (int dz = 0; dz < SUP; dz++) {
                    const int64_t gz = wrap_idx(b2 + dz, n2);
                    const long long addr = in_slab
                        ? (long long)((lx * n1 + gy) * n2 + gz)
                        : (long long)(-1 - lane);
                    merged_deposit(mesh, addr, wxy * w2[dz], lane);
                }
            }
        }
    }
}

// LDS-windowed paint for CELL-SORTED input (TSC/PCS) — each block owns
// a contiguous particle run, accumulates all support^3 deposits into an
// LDS window covering the run's (x, y) line span x the full z axis
// (native ds_add_f64), then flushes the window once with a sequential
// stream of global atomics.  Cuts global atomics from 27 (TSC) / 64
// (PCS) per particle to ~window/run and makes them address-sequential —
// the direct kernel is atomic-issue-bound there (41 vs 127 Gatomic/s
// for CIC's lane-sequential pattern).  Blocks whose particles span a
// window larger than the LDS budget (scrambled input must never reach
// this kernel) fall back to direct per-particle atomics.
#define NBK_TILE_PPB 1024          /* particles per block */
#define NBK_TILE_MAX_CELLS 8192    /* 64 KiB of f64 LDS   */

template <int WINDOW>
__global__ void kpaint_tiled(const double* __restrict__ px,
                             const double* __restrict__ py,
                             const double* __restrict__ pz,
                             const double* __restrict__ mass, int64_t n,
                             int64_t n0, int64_t n1, int64_t n2,
                             double invH0, double invH1, double invH2,
                             double shift,
                             double* __restrict__ mesh,
                             int64_t x0, int64_t nx_local)
{
    constexpr int SUP = (WINDOW == NBK_WINDOW_TSC) ? 3 : 4;
    __shared__ double buf[NBK_TILE_MAX_CELLS];
    __shared__ int s_min0, s_max0, s_min1, s_max1;

    const int64_t ibeg = (int64_t)blockIdx.x * NBK_TILE_PPB;
    const int64_t iend = min(ibeg + NBK_TILE_PPB, n);

    if (threadIdx.x == 0) {
        s_min0 = INT_MAX; s_max0 = INT_MIN;
        s_min1 = INT_MAX; s_max1 = INT_MIN;
    }
    __syncthreads();

    // pass 1: the block's unwrapped base-cell bounding box in (x, y)
    int mn0 = INT_MAX, mx0 = INT_MIN, mn1 = INT_MAX, mx1 = INT_MIN;
    for (int64_t i = ibeg + threadIdx.x; i < iend; i += blockDim.x) {
        const double u0 = px[i] * invH0 + shift;
        const double u1 = py[i] * invH1 + shift;
        int b0, b1;
        if (WINDOW == NBK_WINDOW_TSC) {
            b0 = (int)floor(u0 + 0.5) - 1;
            b1 = (int)floor(u1 + 0.5) - 1;
        } else {
            b0 = (int)floor(u0) - 1;
            b1 = (int)floor(u1) - 1;
        }
        mn0 = min(mn0, b0); mx0 = max(mx0, b0);
        mn1 = min(mn1, b1); mx1 = max(mx1, b1);
    }
    atomicMin(&s_min0, mn0); atomicMax(&s_max0, mx0);
    atomicMin(&s_min1, mn1); atomicMax(&s_max1, mx1);
    __syncthreads();

    const int wx = s_max0 - s_min0 + SUP;
    const int wy = s_max1 - s_min1 + SUP;
    const int64_t cells = (int64_t)wx * wy * n2;
    const bool use_lds = (ibeg < iend) && cells > 0
        && cells <= NBK_TILE_MAX_CELLS;

    if (!use_lds) {
        // fallback: direct per-particle atomics (rare for sorted input)
        for (int64_t i = ibeg + threadIdx.x; i < iend; i += blockDim.x) {
            const double u0 = px[i] * invH0 + shift;
            const double u1 = py[i] * invH1 + shift;
            const double u2 = pz[i] * invH2 + shift;
            const double m = mass ? mass[i] : 1.0;
            double w0[SUP], w1[SUP], w2[SUP];
            int64_t b0, b1, b2;
            paint_weights<WINDOW, SUP>(u0, u1, u2, w0, w1, w2, b0, b1, b2);
            for (int dx = 0; dx < SUP; dx++) {
                const int64_t gx = wrap_idx(b0 + dx, n0);
                if (gx < x0 || gx >= x0 + nx_local) continue;
                for (int dy = 0; dy < SUP; dy++) {
                    const int64_t gy = wrap_idx(b1 + dy, n1);
                    const double wxy = w0[dx] * w1[dy] * m;
                    for (int dz = 0; dz < SUP; dz++) {
                        const int64_t gz = wrap_idx(b2 + dz, n2);
                        atomicAdd(&mesh[((gx - x0) * n1 + gy) * n2 + gz],
                                  wxy * w2[dz]);
                    }
                }
            }
        }
        return;
    }

    for (int64_t w = threadIdx.x; w < cells; w += blockDim.x)
        buf[w] = 0.0;
    __syncthreads();

    // pass 2: deposit into the LDS window
    for (int64_t i = ibeg + threadIdx.x; i < iend; i += blockDim.x) {
        const double u0 = px[i] * invH0 + shift;
        const double u1 = py[i] * invH1 + shift;
        const double u2 = pz[i] * invH2 + shift;
        const double m = mass ? mass[i] : 1.0;
        double w0[SUP], w1[SUP], w2[SUP];
        int64_t b0, b1, b2;
        paint_weights<WINDOW, SUP>(u0, u1, u2, w0, w1, w2, b0, b1, b2);
        const int lx = (int)(b0 - s_min0);
        const int ly = (int)(b1 - s_min1);
        for (int dx = 0; dx < SUP; dx++) {
            for (int dy = 0; dy < SUP; dy++) {
                const double wxy = w0[dx] * w1[dy] * m;
                const int64_t base =
                    ((int64_t)(lx + dx) * wy + (ly + dy)) * n2;
                for (int dz = 0; dz < SUP; dz++) {
                    const int64_t gz = wrap_idx(b2 + dz, n2);
                    unsafeAtomicAdd(&buf[base + gz], wxy * w2[dz]);
                }
            }
        }
    }
    __syncthreads();

    // pass 3: flush (sequential global atomics; skip empty cells)
    for (int64_t w = threadIdx.x; w < cells; w += blockDim.x) {
        const double v = buf[w];
        if (v == 0.0) continue;
        const int64_t gz = w % n2;
        const int64_t wyidx = (w / n2) % wy;
        const int64_t wxidx = w / ((int64_t)n2 * wy);
        const int64_t gx = wrap_idx(s_min0 + wxidx, n0);
        if (gx < x0 || gx >= x0 + nx_local) continue;
        const int64_t gy = wrap_idx(s_min1 + wyidx, n1);
        atomicAdd(&mesh[((gx - x0) * n1 + gy) * n2 + gz], v);
    }
}

// OWNERSHIP-GATHER paint for cell-sorted input with a row table
// (nbk_bucket_fine_f64's rowtab): each block owns an exclusive mesh
// tile of RG y-rows x full z of ONE x-plane, accumulates every deposit
// that lands in its tile from the (few) source rows whose stencils can
// reach it, and flushes with PLAIN stores — zero global atomics.  The
// global atomic pipe measures ~25 G ops/s (csrc/count_probe.hip), which
// bounds the scatter kernels above at ~8 atomics/particle; here the
// deposits are LDS ds_add_f64 and the HBM traffic is ~2.1x the particle
// reads plus one mesh write.
// DOFFT: after the deposits the tile's rows (full z-lines) are
// transformed in place (packed-real radix-2, identical math to
// kfft_r2c_z) and the z half-spectrum is written directly — the real
// mesh never touches HBM.  The row stride is padded (+4 doubles) so the
// butterflies of different rows land in different LDS banks.
// PT: compile-time plane count when > 0 (PT=1 restores the
// single-plane kernel's indexing with zero multi-plane overhead — the
// CIC path is VALU+read balanced and pays for every extra op);
// PT=0 = runtime P.
template <int WINDOW, bool DOFFT, int PT>
__global__ void kpaint_gather(const double* __restrict__ px,
                              const double* __restrict__ py,
                              const double* __restrict__ pz,
                              const double* __restrict__ mass, int64_t n,
                              int64_t n0, int64_t n1, int64_t n2,
                              double invH0, double invH1, double invH2,
                              double shift,
                              const int* __restrict__ rowtab,
                              double* __restrict__ mesh, /* or z-spectrum
                                  (nx_local, n1, n2/2+1) cdouble when
                                  DOFFT */
                              int64_t x0, int64_t nx_local,
                              int RG, int P, int xlo, int xhi,
                              int accumulate,
                              const cdouble* __restrict__ table /* W_n2 */,
                              double scale,
                              int gs, /* >= 0: rowtab is the PAIR-BUCKET
                                  table [(n0/2)*(n1>>gs)+1] of the
                                  duplicating sort (1<<gs == RG); the
                                  tile reads whole (pair, group) ranges
                                  and the deposit masks drop the
                                  out-of-tile copies.  -1: per-row
                                  table (nbk_bucket_fine_f64) */
                              int phases /* perf decomposition only
                                  (NBK_PAINT_PHASES): bit1 deposits,
                                  bit2 FFT+flush; 3 = real kernel */)
{
    constexpr int SUP = (WINDOW == NBK_WINDOW_CIC) ? 2
                      : (WINDOW == NBK_WINDOW_TSC) ? 3 : 4;
    if (PT > 0) P = PT;                   // compile-time plane count
    extern __shared__ double tile[];      // P * RG * (n2 [+4]) doubles
    const int64_t sp = DOFFT ? n2 + 4 : n2;   // padded row stride
    const int64_t tiles_per_plane = n1 / RG;
    // block owns P consecutive x-planes x RG y-rows: a source plane
    // feeding several owned planes is READ ONCE for all of them, so the
    // per-particle read factor drops from (1 + span) planes to
    // (P + span)/P — the host picks P ~ RG to balance the two halo
    // factors within the 160 KiB LDS budget
    const int64_t px0 = x0 + (blockIdx.x / tiles_per_plane) * P;
    const int64_t r0 = (blockIdx.x % tiles_per_plane) * RG;
    const int T = blockDim.x;
    const int t = threadIdx.x;
    const int64_t win = (int64_t)P * RG * sp;

    for (int64_t w = t; w < win; w += T) tile[w] = 0.0;
    __syncthreads();

    // row intervals (wrapped) whose particles can deposit into the tile
    const int64_t rspan = (int64_t)RG + (xhi - xlo);
    int64_t prev_pair = -1;
    for (int dp = xlo; dp <= P - 1 + xhi; dp++) {
        const int64_t p = wrap_idx(px0 + dp, n0);
        int64_t ivals[2][2];
        int niv;
        if (gs >= 0) {
            // PAIR-BUCKET mode: the duplicating sort already placed a
            // copy of every stencil-relevant particle in this tile's
            // own (pair, group) bucket — one contiguous range, no
            // halo-row lookups.  Consecutive source planes share a
            // pair: read each pair once.
            const int64_t q = p >> 1;
            if (q == prev_pair) continue;
            prev_pair = q;
            const int64_t ng = n1 >> gs;
            const int64_t bidx = q * ng + (r0 >> gs);
            ivals[0][0] = rowtab[bidx];
            ivals[0][1] = rowtab[bidx + 1];
            niv = 1;
        } else if (rspan >= n1) {
            ivals[0][0] = 0; ivals[0][1] = n1 - 1; niv = 1;
        } else {
            const int64_t a = wrap_idx(r0 + xlo, n1);
            const int64_t b = wrap_idx(r0 + RG - 1 + xhi, n1);
            if (a <= b) { ivals[0][0] = a; ivals[0][1] = b; niv = 1; }
            else {
                ivals[0][0] = a; ivals[0][1] = n1 - 1;
                ivals[1][0] = 0; ivals[1][1] = b; niv = 2;
            }
        }
        for (int v = 0; v < niv; v++) {
            const int64_t i0 = (gs >= 0) ? ivals[v][0]
                : rowtab[p * n1 + ivals[v][0]];
            const int64_t i1 = (gs >= 0) ? ivals[v][1]
                : rowtab[p * n1 + ivals[v][1] + 1];
            for (int64_t i = i0 + t; i < i1; i += T) {
                const double u0 = px[i] * invH0 + shift;
                const double u1 = py[i] * invH1 + shift;
                const double u2 = pz[i] * invH2 + shift;
                const double m = mass ? mass[i] : 1.0;
                if (!(phases & 1)) {
                    // reads-only decomposition mode: keep the loads
                    // observable, skip the deposit work
                    if (u0 + u1 + u2 + m == 1e308) tile[0] += 1.0;
                    continue;
                }
                double w0[SUP], w1[SUP], w2[SUP];
                int64_t b0, b1, b2;
                paint_weights<WINDOW, SUP>(u0, u1, u2, w0, w1, w2,
                                           b0, b1, b2);
                #pragma unroll
                for (int dx = 0; dx < SUP; dx++) {
                    int64_t pl;
                    if (PT == 1) {
                        if (wrap_idx(b0 + dx, n0) != px0) continue;
                        pl = 0;
                    } else {
                        pl = wrap_idx(b0 + dx, n0) - px0;
                        if (pl < 0) pl += n0;
                        if (pl >= P) continue;
                    }
                    #pragma unroll
                    for (int dy = 0; dy < SUP; dy++) {
                        int64_t ly = wrap_idx(b1 + dy, n1) - r0;
                        if (ly < 0) ly += n1;
                        if (ly >= RG) continue;
                        const double wxy = w0[dx] * w1[dy] * m;
                        #pragma unroll
                        for (int dz = 0; dz < SUP; dz++) {
                            const int64_t gz = wrap_idx(b2 + dz, n2);
                            unsafeAtomicAdd(
                                &tile[(pl * RG + ly) * sp + gz],
                                wxy * w2[dz]);
                        }
                    }
                }
            }
        }
    }
    __syncthreads();

    if (!(phases & 2)) return;       // decomposition mode: no flush/FFT

    if (!DOFFT) {
        // flush the exclusively-owned tile with plain stores
        if (accumulate) {
            for (int64_t w = t; w < win; w += T) {
                const int64_t pl = (PT == 1) ? 0
                    : w / ((int64_t)RG * sp);
                const int64_t rem = (PT == 1) ? w : w - pl * RG * sp;
                const int64_t r = rem / sp, z = rem - r * sp;
                if (z < n2 && tile[w] != 0.0)
                    mesh[((px0 - x0 + pl) * n1 + r0 + r) * n2 + z]
                        += tile[w];
            }
        } else {
            for (int64_t w = t; w < win; w += T) {
                const int64_t pl = (PT == 1) ? 0
                    : w / ((int64_t)RG * sp);
                const int64_t rem = (PT == 1) ? w : w - pl * RG * sp;
                const int64_t r = rem / sp, z = rem - r * sp;
                if (z < n2)
                    mesh[((px0 - x0 + pl) * n1 + r0 + r) * n2 + z]
                        = tile[w];
            }
        }
        return;
    }

    // ---- fused forward z-FFT, one WAVE per tile row (math identical
    // to kfft_r2c_z: packed-real radix-2 DIT + untwiddle split).  A
    // wave owns a whole row, so the stage ordering needs no block
    // syncs — CDNA wave64 executes in lockstep and LDS ops complete in
    // program order; the wave_barrier only pins the compiler's
    // scheduling across the cross-lane dependences. ----
    const int m = (int)(n2 >> 1);
    const int bits = 31 - __clz((unsigned)m);
    const int lane = t & 63;
    const int wave = t >> 6;
    const int nw = T >> 6;

    for (int r = wave; r < P * RG; r += nw) {
        const int64_t pl = (PT == 1) ? 0 : r / RG;
        const int64_t rr = (PT == 1) ? r : r - pl * RG;
        cdouble* out = (cdouble*)mesh
            + ((px0 - x0 + pl) * n1 + r0 + rr) * (m + 1);
        cdouble* z = (cdouble*)&tile[(int64_t)r * sp];

        for (int j = lane; j < m; j += 64) {
            const int jr = nbk_bitrev(j, bits);
            if (j < jr) {
                const cdouble a = z[j];
                z[j] = z[jr];
                z[jr] = a;
            }
        }
        __builtin_amdgcn_wave_barrier();

        // fused radix-4 stages (same bit-reversed order; pairs of
        // radix-2 stages become one 4-point butterfly — half the LDS
        // round trips; algebra as lds_fft4 in nbk_fft.hip, validated
        // element-exact against numpy).  Odd log2(m): one multiply-free
        // radix-2 stage first.
        {
            const int fbits = 31 - __clz((unsigned)m);
            int len = 2;
            if (fbits & 1) {
                for (int q = lane; q < (m >> 1); q += 64) {
                    const int i0 = 2 * q;
                    const cdouble u = z[i0];
                    const cdouble v = z[i0 + 1];
                    z[i0] = cadd(u, v);
                    z[i0 + 1] = csub(u, v);
                }
                __builtin_amdgcn_wave_barrier();
                len = 4;
            }
            for (; 2 * len <= m; len <<= 2) {
                const int h = len >> 1;
                const int tw = m / len;
                for (int q = lane; q < (m >> 2); q += 64) {
                    const int grp = q / h;
                    const int pos = q - grp * h;
                    const int base = grp * (len << 1) + pos;
                    const cdouble w1 = table[pos * tw];
                    const cdouble w2 = table[2 * pos * tw];
                    const cdouble w3 = cmul(w1, w2);
                    const cdouble x0 = z[base];
                    const cdouble x1 = z[base + h];
                    const cdouble x2 = z[base + 2 * h];
                    const cdouble x3 = z[base + 3 * h];
                    const cdouble b1 = cmul(x2, w1);
                    const cdouble b2 = cmul(x1, w2);
                    const cdouble b3 = cmul(x3, w3);
                    const cdouble e0 = cadd(x0, b2);
                    const cdouble e1 = csub(x0, b2);
                    const cdouble o0 = cadd(b1, b3);
                    const cdouble o1 = csub(b1, b3);
                    z[base] = cadd(e0, o0);
                    z[base + 2 * h] = csub(e0, o0);
                    const cdouble io1 = {o1.im, -o1.re};   // -i*o1
                    z[base + h] = cadd(e1, io1);
                    z[base + 3 * h] = csub(e1, io1);
                }
                __builtin_amdgcn_wave_barrier();
            }
        }

        // untwiddle split: X[k] = E[k] + W_n2^k O[k], k = 0..m, written
        // straight to the z half-spectrum
        for (int k = lane; k <= m; k += 64) {
            const cdouble Zk = z[k == m ? 0 : k];
            const cdouble Zm = z[(m - k) % m];
            const cdouble E = cscale(cadd(Zk, cconj(Zm)), 0.5);
            const cdouble D = csub(Zk, cconj(Zm));
            const cdouble O = {0.5 * D.im, -0.5 * D.re};  // D * (-i/2)
            const cdouble X = cadd(E, cmul(table[k], O));
            out[k] = cscale(X, scale);
        }
        __builtin_amdgcn_wave_barrier();
    }
}

// windowed gather — the dual of kpaint (pmesh readout, used by
// FFTRecon's displacement solve, fftrecon.py:246-249, and the nnb
// variant by the LogNormal generator, mockmaker.py:317-319).  Each lane
// sums W * mesh over its particle's neighbourhood; cells outside the
// local slab contribute 0 (the ghost owner adds its partial — the host
// exchanges and sums partials across ranks).
template <int WINDOW>
__global__ void kreadout(const double* __restrict__ px,
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
        const double u0 = px[i] * invH0, u1 = py[i] * invH1,
                     u2 = pz[i] * invH2;

        if (WINDOW == 3) {    // nnb
            const int64_t gx = wrap_idx((int64_t)floor(u0 + 0.5), n0);
            const int64_t gy = wrap_idx((int64_t)floor(u1 + 0.5), n1);
            const int64_t gz = wrap_idx((int64_t)floor(u2 + 0.5), n2);
            out[i] = (gx >= x0 && gx < x0 + nx_local)
                ? mesh[((gx - x0) * n1 + gy) * n2 + gz] : 0.0;
            continue;
        }

        constexpr int SUP = (WINDOW == NBK_WINDOW_CIC) ? 2
                          : (WINDOW == NBK_WINDOW_TSC) ? 3 : 4;
        double w0[SUP], w1[SUP], w2[SUP];
        int64_t b0, b1, b2;
        if (WINDOW == NBK_WINDOW_CIC) {
            const double f0 = floor(u0), f1 = floor(u1), f2 = floor(u2);
            b0 = (int64_t)f0; b1 = (int64_t)f1; b2 = (int64_t)f2;
            w0[1] = u0 - f0; w0[0] = 1.0 - w0[1];
            w1[1] = u1 - f1; w1[0] = 1.0 - w1[1];
            w2[1] = u2 - f2; w2[0] = 1.0 - w2[1];
        } else if (WINDOW == NBK_WINDOW_TSC) {
            const double f0 = floor(u0 + 0.5), f1 = floor(u1 + 0.5),
                         f2 = floor(u2 + 0.5);
            b0 = (int64_t)f0 - 1; b1 = (int64_t)f1 - 1;
            b2 = (int64_t)f2 - 1;
            #pragma unroll
            for (int d = 0; d < 3; d++) {
                const double s0 = u0 - (f0 + d - 1);
                const double s1 = u1 - (f1 + d - 1);
                const double s2 = u2 - (f2 + d - 1);
                const double a0 = fabs(s0), a1 = fabs(s1), a2 = fabs(s2);
                w0[d] = a0 < 0.5 ? 0.75 - s0 * s0
                                 : 0.5 * (1.5 - a0) * (1.5 - a0);
                w1[d] = a1 < 0.5 ? 0.75 - s1 * s1
                                 : 0.5 * (1.5 - a1) * (1.5 - a1);
                w2[d] = a2 < 0.5 ? 0.75 - s2 * s2
                                 : 0.5 * (1.5 - a2) * (1.5 - a2);
            }
        } else {
            const double f0 = floor(u0), f1 = floor(u1), f2 = floor(u2);
            b0 = (int64_t)f0 - 1; b1 = (int64_t)f1 - 1;
            b2 = (int64_t)f2 - 1;
            #pragma unroll
            for (int d = 0; d < 4; d++) {
                const double s0 = fabs(u0 - (f0 + d - 1));
                const double s1 = fabs(u1 - (f1 + d - 1));
                const double s2 = fabs(u2 - (f2 + d - 1));
                w0[d] = s0 < 1.0
                    ? (4.0 - 6.0 * s0 * s0 + 3.0 * s0 * s0 * s0) / 6.0
                    : (2.0 - s0) * (2.0 - s0) * (2.0 - s0) / 6.0;
                w1[d] = s1 < 1.0
                    ? (4.0 - 6.0 * s1 * s1 + 3.0 * s1 * s1 * s1) / 6.0
                    : (2.0 - s1) * (2.0 - s1) * (2.0 - s1) / 6.0;
                w2[d] = s2 < 1.0
                    ? (4.0 - 6.0 * s2 * s2 + 3.0 * s2 * s2 * s2) / 6.0
                    : (2.0 - s2) * (2.0 - s2) * (2.0 - s2) / 6.0;
            }
        }

        double acc = 0.0;
        #pragma unroll
        for (int dx = 0; dx < SUP; dx++) {
            const int64_t gx = wrap_idx(b0 + dx, n0);
            if (gx < x0 || gx >= x0 + nx_local) continue;
            const int64_t lx = gx - x0;
            #pragma unroll
            for (int dy = 0; dy < SUP; dy++) {
                const int64_t gy = wrap_idx(b1 + dy, n1);
                const double wxy = w0[dx] * w1[dy];
                #pragma unroll
                for (int dz = 0; dz < SUP; dz++) {
                    const int64_t gz = wrap_idx(b2 + dz, n2);
                    acc += wxy * w2[dz]
                        * mesh[(lx * n1 + gy) * n2 + gz];
                }
            }
        }
        out[i] = acc;
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
    const int grid = grid_for(n, block);
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

extern "C" int nbk_paint_sorted_f64(const double* pos, const double* mass,
                                    int64_t n, const int64_t nmesh[3],
                                    const double box[3], int window,
                                    double shift, double* mesh, int64_t x0,
                                    int64_t nx_local, void* stream)
{
    // Measured on the C3 input (1e8 TSC, bucket-sorted): the LDS-window
    // path is 76 vs 45 ms against the wave-merged direct kernel — PMC
    // shows all TSC variants 71-88% instruction-ISSUE-stalled, so
    // rerouting deposits through LDS buys nothing there.  Only PCS
    // (64 deposits, flush ~3x cheaper than direct) takes the tiled
    // path; CIC and TSC keep the direct kernel.
    if (window == NBK_WINDOW_CIC || window == NBK_WINDOW_TSC || n == 0)
        return nbk_paint_f64(pos, mass, n, nmesh, box, window, shift,
                             mesh, x0, nx_local, stream);
    if (window != NBK_WINDOW_TSC && window != NBK_WINDOW_PCS) {
        NBK_SET_ERR("nbk_paint_sorted_f64: unknown window id %d", window);
        return NBK_ERR_ARG;
    }
    const double invH0 = nmesh[0] / box[0];
    const double invH1 = nmesh[1] / box[1];
    const double invH2 = nmesh[2] / box[2];
    const int64_t grid = (n + NBK_TILE_PPB - 1) / NBK_TILE_PPB;
    if (grid > 0x7fffffff) {
        NBK_SET_ERR("nbk_paint_sorted_f64: grid too large");
        return NBK_ERR_ARG;
    }
    hipStream_t s = (hipStream_t)stream;
    const double *px = pos, *py = pos + n, *pz = pos + 2 * n;
    if (window == NBK_WINDOW_TSC)
        hipLaunchKernelGGL(kpaint_tiled<NBK_WINDOW_TSC>,
                           dim3((uint32_t)grid), dim3(256), 0, s,
                           px, py, pz, mass, n, nmesh[0], nmesh[1],
                           nmesh[2], invH0, invH1, invH2, shift, mesh,
                           x0, nx_local);
    else
        hipLaunchKernelGGL(kpaint_tiled<NBK_WINDOW_PCS>,
                           dim3((uint32_t)grid), dim3(256), 0, s,
                           px, py, pz, mass, n, nmesh[0], nmesh[1],
                           nmesh[2], invH0, invH1, invH2, shift, mesh,
                           x0, nx_local);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_readout_f64(const double* pos, int64_t n,
                               const int64_t nmesh[3], const double box[3],
                               int window,
                               const double* mesh, int64_t x0,
                               int64_t nx_local,
                               double* out, void* stream)
{
    if (n == 0) return NBK_OK;
    const int block = 256;
    const int grid = grid_for(n, block);
    hipStream_t s = (hipStream_t)stream;
    const double iH0 = nmesh[0] / box[0];
    const double iH1 = nmesh[1] / box[1];
    const double iH2 = nmesh[2] / box[2];
    switch (window) {
    case NBK_WINDOW_CIC:
        hipLaunchKernelGGL(kreadout<NBK_WINDOW_CIC>, dim3(grid), dim3(block),
                           0, s, pos, pos + n, pos + 2 * n, n,
                           nmesh[0], nmesh[1], nmesh[2], iH0, iH1, iH2,
                           mesh, x0, nx_local, out);
        break;
    case NBK_WINDOW_TSC:
        hipLaunchKernelGGL(kreadout<NBK_WINDOW_TSC>, dim3(grid), dim3(block),
                           0, s, pos, pos + n, pos + 2 * n, n,
                           nmesh[0], nmesh[1], nmesh[2], iH0, iH1, iH2,
                           mesh, x0, nx_local, out);
        break;
    case NBK_WINDOW_PCS:
        hipLaunchKernelGGL(kreadout<NBK_WINDOW_PCS>, dim3(grid), dim3(block),
                           0, s, pos, pos + n, pos + 2 * n, n,
                           nmesh[0], nmesh[1], nmesh[2], iH0, iH1, iH2,
                           mesh, x0, nx_local, out);
        break;
    case 3:   /* nnb */
        hipLaunchKernelGGL(kreadout<3>, dim3(grid), dim3(block), 0, s,
                           pos, pos + n, pos + 2 * n, n,
                           nmesh[0], nmesh[1], nmesh[2], iH0, iH1, iH2,
                           mesh, x0, nx_local, out);
        break;
    default:
        NBK_SET_ERR("nbk_readout_f64: unknown window id %d", window);
        return NBK_ERR_ARG;
    }
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}


// pick the gather tile shape (P x-planes x RG y-rows): minimize the
// particle re-read factor (P+sx)/P * (RG+sy)/RG (sx/sy = source span
// beyond the tile from the window support and interlace shift) within
// the 160 KiB LDS budget; powers of two dividing the local dims
// NBK_PAINT_PHASES: perf decomposition only (1 = reads+deposits,
// 0 = reads only, 2 = flush/FFT on an empty tile); default 3
static int nbk_paint_phases(void) {
    static int v = -1;
    if (v < 0) {
        const char* e = getenv("NBK_PAINT_PHASES");
        v = e ? atoi(e) : 3;
        if (v < 0 || v > 3) v = 3;
    }
    return v;
}

static void nbk_pick_tile(int64_t nx_local, int64_t n1, int64_t n2,
                          int64_t pad, int sx, int sy,
                          int* P_out, int* RG_out)
{
    const int64_t budget = 20480 / (n2 + pad);
    // NBK_PAINT_P forces the plane count (tuning); RG fills the budget
    static int forceP = -1;
    if (forceP < 0) {
        const char* e = getenv("NBK_PAINT_P");
        forceP = e ? atoi(e) : 0;
        if (forceP < 0 || forceP > 16) forceP = 0;
    }
    double best = 1e30;
    int bP = 1, bRG = 1;
    // Measured policy (r02 A/B at C4/C3): narrow-span windows (CIC)
    // run fastest single-plane (the PT=1 specialized kernel: pure
    // paint 13.8 vs 15.7 ms, fused 19.9 vs 20.4 at C4) — the extra
    // source-plane read is cheaper than the multi-plane bookkeeping;
    // wide-span windows (TSC/PCS, 4 source planes single-plane) win
    // with the balanced multi-plane tile (C3 fused 4.8 vs 5.1 ms).
    const int prefer_single = (sx <= 1) && !forceP;
    for (int64_t P = 1; P <= nx_local && P <= budget; P <<= 1) {
        if (nx_local % P) break;
        if (forceP && P != forceP) continue;
        if (prefer_single && P > 1) break;
        for (int64_t RG = 1; RG <= n1 && P * RG <= budget; RG <<= 1) {
            if (n1 % RG) break;
            double cost = (double)(P + sx) / (double)P
                        * (double)(RG + sy) / (double)RG;
            if (forceP || prefer_single)
                cost = 1.0 / (double)RG;           // max RG at this P
            if (cost < best - 1e-12) { best = cost; bP = (int)P; bRG = (int)RG; }
        }
    }
    *P_out = bP;
    *RG_out = bRG;
}

extern "C" int nbk_paint_gather_f64(const double* pos, const double* mass,
                                    int64_t n, const int64_t nmesh[3],
                                    const double box[3],
                                    int window, double shift,
                                    const int* rowtab,
                                    double* mesh, int64_t x0,
                                    int64_t nx_local, int accumulate,
                                    int pair_gs, void* stream)
{
    const int64_t n0 = nmesh[0], n1 = nmesh[1], n2 = nmesh[2];
    if (n2 > 20480) {
        NBK_SET_ERR("nbk_paint_gather_f64: no LDS tile for n1=%lld "
                    "n2=%lld", (long long)n1, (long long)n2);
        return NBK_ERR_UNSUPPORTED;
    }
    // delta = b0 - ixc range per window/shift: source
    // planes/rows = [xlo, xhi] around the owner
    const bool sh = shift != 0.0;
    int sup, dmin, dmax;
    if (window == NBK_WINDOW_CIC) {
        sup = 2; dmin = 0; dmax = sh ? 1 : 0;
    } else if (window == NBK_WINDOW_TSC) {
        sup = 3; dmin = sh ? 0 : -1; dmax = 0;
    } else if (window == NBK_WINDOW_PCS) {
        sup = 4; dmin = -1; dmax = sh ? 0 : -1;
    } else {
        NBK_SET_ERR("nbk_paint_gather_f64: bad window %d", window);
        return NBK_ERR_ARG;
    }
    const int xlo = -sup + 1 - dmax;
    const int xhi = -dmin;
    const int span = xhi - xlo;

    int P, RG;
    if (pair_gs >= 0) {
        // pair-bucket mode: tile geometry is pinned by the sort's
        // group size (one plane x one y-group)
        P = 1;
        RG = 1 << pair_gs;
        if ((int64_t)RG * n2 * 8 > 160 * 1024 || n1 % RG) {
            NBK_SET_ERR("nbk_paint_gather_f64: bad pair_gs=%d", pair_gs);
            return NBK_ERR_ARG;
        }
    } else {
        nbk_pick_tile(nx_local, n1, n2, 0, span, span, &P, &RG);
    }
    const int64_t grid = (nx_local / P) * (n1 / RG);
    const size_t lds = (size_t)P * RG * n2 * sizeof(double);
    hipStream_t s = (hipStream_t)stream;
    if (lds > 64 * 1024) {
        const void* fns[6] = {
            reinterpret_cast<const void*>(
                &kpaint_gather<NBK_WINDOW_CIC, false, 0>),
            reinterpret_cast<const void*>(
                &kpaint_gather<NBK_WINDOW_TSC, false, 0>),
            reinterpret_cast<const void*>(
                &kpaint_gather<NBK_WINDOW_PCS, false, 0>),
            reinterpret_cast<const void*>(
                &kpaint_gather<NBK_WINDOW_CIC, false, 1>),
            reinterpret_cast<const void*>(
                &kpaint_gather<NBK_WINDOW_TSC, false, 1>),
            reinterpret_cast<const void*>(
                &kpaint_gather<NBK_WINDOW_PCS, false, 1>)};
        (void)hipFuncSetAttribute(fns[window + (P == 1 ? 3 : 0)],
            hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
    }
    #define NBK_LAUNCH_GATHER(W, PT) \
        hipLaunchKernelGGL((kpaint_gather<W, false, PT>), \
                           dim3((uint32_t)grid), \
                           dim3(1024), lds, s, pos, pos + n, pos + 2 * n, \
                           mass, n, n0, n1, n2, \
                           n0 / box[0], n1 / box[1], n2 / box[2], shift, \
                           rowtab, mesh, x0, nx_local, RG, P, xlo, xhi, \
                           accumulate, (const cdouble*)nullptr, 1.0, \
                           pair_gs, nbk_paint_phases())
    if (P == 1) {
        if (window == NBK_WINDOW_CIC) NBK_LAUNCH_GATHER(NBK_WINDOW_CIC, 1);
        else if (window == NBK_WINDOW_TSC) NBK_LAUNCH_GATHER(NBK_WINDOW_TSC, 1);
        else NBK_LAUNCH_GATHER(NBK_WINDOW_PCS, 1);
    } else {
        if (window == NBK_WINDOW_CIC) NBK_LAUNCH_GATHER(NBK_WINDOW_CIC, 0);
        else if (window == NBK_WINDOW_TSC) NBK_LAUNCH_GATHER(NBK_WINDOW_TSC, 0);
        else NBK_LAUNCH_GATHER(NBK_WINDOW_PCS, 0);
    }
    #undef NBK_LAUNCH_GATHER
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_paint_gather_fft_f64(const double* pos,
                                        const double* mass, int64_t n,
                                        const int64_t nmesh[3],
                                        const double box[3],
                                        int window, double shift,
                                        const int* rowtab,
                                        double* zspec, int64_t x0,
                                        int64_t nx_local, double scale,
                                        int pair_gs, void* stream)
{
    const int64_t n0 = nmesh[0], n1 = nmesh[1], n2 = nmesh[2];
    if (n2 < 8 || n2 > 4096 || (n2 & (n2 - 1))) {
        NBK_SET_ERR("nbk_paint_gather_fft_f64: n2=%lld not a supported "
                    "FFT length", (long long)n2);
        return NBK_ERR_UNSUPPORTED;
    }
    if (n2 + 4 > 20480) {
        NBK_SET_ERR("nbk_paint_gather_fft_f64: no LDS tile for n1=%lld "
                    "n2=%lld", (long long)n1, (long long)n2);
        return NBK_ERR_UNSUPPORTED;
    }
    const double* table = nbk_internal_twiddles(n2);
    if (!table) {
        NBK_SET_ERR("twiddle alloc failed");
        return NBK_ERR_HIP;
    }
    const bool sh = shift != 0.0;
    int sup, dmin, dmax;
    if (window == NBK_WINDOW_CIC) {
        sup = 2; dmin = 0; dmax = sh ? 1 : 0;
    } else if (window == NBK_WINDOW_TSC) {
        sup = 3; dmin = sh ? 0 : -1; dmax = 0;
    } else if (window == NBK_WINDOW_PCS) {
        sup = 4; dmin = -1; dmax = sh ? 0 : -1;
    } else {
        NBK_SET_ERR("nbk_paint_gather_fft_f64: bad window %d", window);
        return NBK_ERR_ARG;
    }
    const int xlo = -sup + 1 - dmax;
    const int xhi = -dmin;
    const int span = xhi - xlo;

    int P, RG;
    if (pair_gs >= 0) {
        P = 1;
        RG = 1 << pair_gs;
        if ((int64_t)RG * (n2 + 4) * 8 > 160 * 1024 || n1 % RG) {
            NBK_SET_ERR("nbk_paint_gather_fft_f64: bad pair_gs=%d",
                        pair_gs);
            return NBK_ERR_ARG;
        }
    } else {
        nbk_pick_tile(nx_local, n1, n2, 4, span, span, &P, &RG);
    }
    const int64_t grid = (nx_local / P) * (n1 / RG);
    const size_t lds = (size_t)P * RG * (n2 + 4) * sizeof(double);
    hipStream_t s = (hipStream_t)stream;
    if (lds > 64 * 1024) {
        const void* fns[6] = {
            reinterpret_cast<const void*>(
                &kpaint_gather<NBK_WINDOW_CIC, true, 0>),
            reinterpret_cast<const void*>(
                &kpaint_gather<NBK_WINDOW_TSC, true, 0>),
            reinterpret_cast<const void*>(
                &kpaint_gather<NBK_WINDOW_PCS, true, 0>),
            reinterpret_cast<const void*>(
                &kpaint_gather<NBK_WINDOW_CIC, true, 1>),
            reinterpret_cast<const void*>(
                &kpaint_gather<NBK_WINDOW_TSC, true, 1>),
            reinterpret_cast<const void*>(
                &kpaint_gather<NBK_WINDOW_PCS, true, 1>)};
        (void)hipFuncSetAttribute(fns[window + (P == 1 ? 3 : 0)],
            hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
    }
    #define NBK_LAUNCH_GFFT(W, PT) \
        hipLaunchKernelGGL((kpaint_gather<W, true, PT>), \
                           dim3((uint32_t)grid), \
                           dim3(1024), lds, s, pos, pos + n, pos + 2 * n, \
                           mass, n, n0, n1, n2, \
                           n0 / box[0], n1 / box[1], n2 / box[2], shift, \
                           rowtab, zspec, x0, nx_local, RG, P, xlo, \
                           xhi, 0, (const cdouble*)table, scale, \
                           pair_gs, nbk_paint_phases())
    if (P == 1) {
        if (window == NBK_WINDOW_CIC) NBK_LAUNCH_GFFT(NBK_WINDOW_CIC, 1);
        else if (window == NBK_WINDOW_TSC) NBK_LAUNCH_GFFT(NBK_WINDOW_TSC, 1);
        else NBK_LAUNCH_GFFT(NBK_WINDOW_PCS, 1);
    } else {
        if (window == NBK_WINDOW_CIC) NBK_LAUNCH_GFFT(NBK_WINDOW_CIC, 0);
        else if (window == NBK_WINDOW_TSC) NBK_LAUNCH_GFFT(NBK_WINDOW_TSC, 0);
        else NBK_LAUNCH_GFFT(NBK_WINDOW_PCS, 0);
    }
    #undef NBK_LAUNCH_GFFT
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}
