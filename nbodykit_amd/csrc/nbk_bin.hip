// project_to_basis binning sums (nbodykit/algorithms/fftpower.py:507-701
// with the SlabIterator/Hermitian-weight semantics of meshtools.py):
// one streaming pass over the local complex slab, digitizing |k|^2 and
// mu against the edge arrays and accumulating
//   xsum  += |k| * w,  musum += mu * w,  Nsum += w,
//   ysum[ell] += (2 ell + 1) * Herm(P_ell(mu) * p3d)
// into (Nx+2)x(Nmu+2) bin grids, where w is the Hermitian double-count
// weight (2 where the compressed-axis frequency > 0, meshtools:188-215)
// and Herm applies the conjugate-pair parity rule (fftpower.py:649-656).
//
// Fused mode (nbk_power_bin_f64) computes comp1(c1) conj(comp2(c2)) V
// per element on the fly — compensation, cross power and binning in
// one pass, p3d never materialized — and skips the ~48% of modes
// beyond the last k-edge before touching field data.  Line-grouped
// block mapping keeps the expensive index math per line, with the
// digitize edges staged through LDS.  Histograms accumulate in LDS
// (up to 160 KiB via hipFuncSetAttribute) and flush once; the
// global-atomic fallback serves grids beyond that.
#include "nbk_common.h"

namespace {

#define NBK_MAX_ELL 8

struct BinArgs {
    int64_t n0, n1, n2;        // global REAL mesh (n2 = full length)
    int64_t d0, d1, d2;        // local dims (d2 = n2/2+1 when untransposed)
    int64_t o0, o1, o2;
    int a0, a1, a2;
    double k0x, k0y, k0z;      // 2 pi / BoxSize
    double losx, losy, losz;
    int nx_edges, nmu_edges;   // edge COUNTS (bins + 1)
    int nell;
    int real_field;            // 1: y3d is a REAL configuration-space
                               // field (FFTCorr, fftcorr.py:150-176):
                               // coords are relative positions, all axes
                               // full-length, no Hermitian double-count
    int ells[NBK_MAX_ELL];     // multipole orders, by value (kernel args
                               // live in SGPRs — no host pointer deref)
    // fused compensate+power mode (nbk_power_bin_f64): the element value
    // is comp1(c1) * conj(comp2(c2)) * volume computed on the fly instead
    // of a pre-materialized p3d — one read pass replaces the
    // compensate/power3d/bin trio (window -1 = no compensation)
    int fuse;
    int win1, interl1, win2, interl2;
    int clear_zero;
    double volume;
};

// numpy.digitize(x, edges) == count of edges <= x (right-open bins)
__device__ __forceinline__ int dig(const double* __restrict__ edges,
                                   int nedges, double x) {
    int lo = 0, hi = nedges;
    while (lo < hi) {
        const int mid = (lo + hi) >> 1;
        if (edges[mid] <= x) lo = mid + 1;
        else hi = mid;
    }
    return lo;
}

// accumulate one element into hist arrays laid out as:
//   [xsum (NB)] [musum (NB)] [Nsum (NB)] [ysum: nell * 2 * NB]
// where NB = (Nx+2)*(Nmu+2)
template <bool LDS>
__device__ __forceinline__ void accum(double* __restrict__ h, int NB,
                                      int bin, double kmag, double mu,
                                      double w, const cdouble* yv, int nell) {
    atomicAdd(&h[bin], kmag * w);
    atomicAdd(&h[NB + bin], mu * w);
    atomicAdd(&h[2 * NB + bin], w);
    for (int e = 0; e < nell; e++) {
        atomicAdd(&h[(3 + 2 * e) * NB + bin], yv[e].re);
        atomicAdd(&h[(3 + 2 * e + 1) * NB + bin], yv[e].im);
    }
}

// accumulate pre-merged register sums for one bin (the run-merging fast
// path flushes these on bin changes instead of per element)
template <bool LDS>
__device__ __forceinline__ void accum_sums(double* __restrict__ h, int NB,
                                           int bin, double ak, double amu,
                                           double aw, const cdouble* ay,
                                           int nell) {
    atomicAdd(&h[bin], ak);
    atomicAdd(&h[NB + bin], amu);
    atomicAdd(&h[2 * NB + bin], aw);
    for (int e = 0; e < nell; e++) {
        atomicAdd(&h[(3 + 2 * e) * NB + bin], ay[e].re);
        atomicAdd(&h[(3 + 2 * e + 1) * NB + bin], ay[e].im);
    }
}

// integer frequency of global index g on global axis `axis` (the
// compressed half-spectrum convention applies to global axis 2 of a
// complex field only)
__device__ __forceinline__ double freq_axis(int axis, int64_t g, int64_t n,
                                            int real_field) {
    return (axis == 2 && !real_field) ? freq_half(g, n) : freq_full(g, n);
}

template <bool LDS>
__global__ void kbin(const double* __restrict__ data,
                     const double* __restrict__ data2, BinArgs A,
                     const double* __restrict__ k2edges_g,
                     const double* __restrict__ muedges_g,
                     double* __restrict__ gout /* nfields * NB */)
{
    const int NB = (A.nx_edges + 1) * (A.nmu_edges + 1);
    const int nfields = 3 + 2 * A.nell;

    // LDS: [histograms | k2 edges | mu edges] — the digitize binary
    // search runs against LDS instead of chasing L1/L2 lines
    extern __shared__ double lh[];
    double* h = gout;
    const double* k2edges = k2edges_g;
    const double* muedges = muedges_g;
    if (LDS) {
        for (int i = threadIdx.x; i < NB * nfields; i += blockDim.x)
            lh[i] = 0.0;
        double* ke = lh + NB * nfields;
        double* me = ke + A.nx_edges;
        for (int i = threadIdx.x; i < A.nx_edges; i += blockDim.x)
            ke[i] = k2edges_g[i];
        for (int i = threadIdx.x; i < A.nmu_edges; i += blockDim.x)
            me[i] = muedges_g[i];
        __syncthreads();
        h = lh;
        k2edges = ke;
        muedges = me;
    }

    // NL z-lines per block, flattened so every thread stays busy across
    // line boundaries (a single line of n2/2+1 elements would leave the
    // tail wave idle); the expensive int64 div/mod runs NL times per
    // GROUP, cooperatively, with the per-line frequencies staged through
    // a small static LDS array
    const int64_t n3[3] = {A.n0, A.n1, A.n2};
    const int64_t nlines = A.d0 * A.d1;
    constexpr int NL = 8;
    __shared__ double lf0[NL], lf1[NL];
    const int64_t ngroups = (nlines + NL - 1) / NL;
    for (int64_t grp = blockIdx.x; grp < ngroups; grp += gridDim.x) {
        const int64_t line0 = grp * NL;
        __syncthreads();
        if (threadIdx.x < NL && line0 + threadIdx.x < nlines) {
            const int64_t line = line0 + threadIdx.x;
            const int64_t l0 = line / A.d1;
            const int64_t l1 = line - l0 * A.d1;
            lf0[threadIdx.x] = freq_axis(A.a0, l0 + A.o0, n3[A.a0],
                                         A.real_field);
            lf1[threadIdx.x] = freq_axis(A.a1, l1 + A.o1, n3[A.a1],
                                         A.real_field);
        }
        __syncthreads();
        const int64_t nelem =
            (nlines - line0 < NL ? nlines - line0 : (int64_t)NL) * A.d2;
        int sub = 0;
        int64_t l2 = threadIdx.x;
        while (l2 >= A.d2) { l2 -= A.d2; sub++; }

    for (int64_t i = threadIdx.x; i < nelem;
         i += blockDim.x,
         l2 += blockDim.x,
         ({ while (l2 >= A.d2) { l2 -= A.d2; sub++; } })) {
        const int64_t idx = (line0 + sub) * A.d2 + l2;
        double f3[3];
        f3[A.a0] = lf0[sub];
        f3[A.a1] = lf1[sub];
        f3[A.a2] = freq_axis(A.a2, l2 + A.o2, n3[A.a2], A.real_field);
        const double fx = f3[0];
        const double fy = f3[1];
        const double fz = f3[2];

        double kx = fx * A.k0x, ky = fy * A.k0y, kz = fz * A.k0z;
        if (A.real_field) {
            // fl(fl(f*L)/N): the Python/oracle real-coordinate rounding
            kx /= (double)A.n0;
            ky /= (double)A.n1;
            kz /= (double)A.n2;
        }
        const double k2 = kx * kx + ky * ky + kz * kz;
        // modes beyond the last edge land in the (Nx+1) overflow row,
        // which project_to_basis discards (fftpower.py:666-668 keeps
        // bins 1..Nx) — skip them before touching the field data.  At
        // default edges (kmax ~ the min-axis Nyquist) this is every
        // corner mode outside the inscribed sphere, ~48% of the volume.
        if (k2 >= k2edges[A.nx_edges - 1]) continue;
        const double kmag = sqrt(k2);
        double mu = kx * A.losx + ky * A.losy + kz * A.losz;
        mu = (kmag == 0.0) ? 0.0 : mu / kmag;

        const bool nonsingular = !A.real_field && fz > 0.0;  // doubled
        const double w = nonsingular ? 2.0 : 1.0;

        const int bx = dig(k2edges, A.nx_edges, k2);
        const int bmu = dig(muedges, A.nmu_edges, mu);
        const int bin = bx * (A.nmu_edges + 1) + bmu;

        cdouble v;
        if (A.fuse) {
            // comp1(c1) * conj(comp2(c2)) * V, zero mode cleared but
            // still binned (fftpower.py:114-128); compensation factors
            // bit-identical to the standalone nbk_compensate_f64 pass
            const double w[3] = {2.0 * M_PI * fx / (double)A.n0,
                                 2.0 * M_PI * fy / (double)A.n1,
                                 2.0 * M_PI * fz / (double)A.n2};
            cdouble a = {data[2 * idx], data[2 * idx + 1]};
            cdouble b = (data2 == data) ? a
                : cdouble{data2[2 * idx], data2[2 * idx + 1]};
            const bool same_comp = (A.win2 == A.win1
                                    && A.interl2 == A.interl1);
            double fac1 = 1.0;
            if (A.win1 >= 0) {
                fac1 = nbk_comp_factor(A.win1, A.interl1, w);
                a = cscale(a, fac1);
            }
            if (A.win2 >= 0) {
                if (data2 == data && same_comp)
                    b = a;
                else
                    b = cscale(b, same_comp && A.win1 >= 0 ? fac1
                               : nbk_comp_factor(A.win2, A.interl2, w));
            }
            cdouble p = cmul(a, cconj(b));
            v = {p.re * A.volume, p.im * A.volume};
            if (A.clear_zero && fx == 0.0 && fy == 0.0 && fz == 0.0)
                v = {0.0, 0.0};
        } else {
            v = A.real_field
                ? cdouble{data[idx], 0.0}
                : cdouble{data[2 * idx], data[2 * idx + 1]};
        }

        // Legendre P_ell(mu) by recurrence, ells ascending with ells[0]==0
        cdouble yv[NBK_MAX_ELL];
        double Pm1 = 0.0, P = 1.0;   // P_{-1}, P_0
        int e = 0;
        for (int l = 0; e < A.nell; l++) {
            if (l > 0) {
                const double Pn = ((2 * l - 1) * mu * P - (l - 1) * Pm1) / l;
                Pm1 = P;
                P = Pn;
            }
            if (l == A.ells[e]) {
                cdouble wy = cscale(v, P);
                // conjugate-pair parity (fftpower.py:649-656)
                if (nonsingular) {
                    if (l % 2) wy = {0.0, 2.0 * wy.im};
                    else wy = {2.0 * wy.re, 0.0};
                }
                yv[e] = cscale(wy, 2.0 * l + 1.0);
                e++;
            }
        }

        accum<LDS>(h, NB, bin, kmag, mu, w, yv, A.nell);
    }
    }

    if (LDS) {
        __syncthreads();
        for (int i = threadIdx.x; i < NB * nfields; i += blockDim.x)
            if (lh[i] != 0.0) atomicAdd(&gout[i], lh[i]);
    }
}

// Fast path for the identity layout of a COMPLEX field (a2 == 2 —
// every call today: axis_map is never non-null).  The general kernel
// above is VALU/LDS-atomic bound (~0.4 TB/s at C4: 3 f64 sins +
// 3 sqrts + 3 divides of the compensation, ~9-probe digitize and
// 3+2*nell LDS atomics PER ELEMENT).  Here:
//  - per-LINE invariants (kx^2+ky^2, kx.losx+ky.losy, the xy part of
//    the separable compensation) are hoisted into the NL-line stage;
//  - the z-axis compensation factors come from a d2-entry LDS table;
//  - each thread walks CONTIGUOUS runs of R elements, merging the
//    histogram contributions of same-bin neighbours in registers
//    (along z the shell index changes slowly) and warm-starting the
//    digitize from the previous bin.
// Bit-parity notes: k2 = (kx*kx + ky*ky) + kz*kz and
// mu = ((kx lx + ky ly) + kz lz)/|k| keep the general kernel's exact
// groupings, so bin assignment is bit-identical; the compensation is
// composed as products of per-axis reciprocals instead of chained
// divides — a last-ulp value difference covered by the 1e-10
// fused-vs-unfused tests.
template <bool LDS>
__global__ void kbin_run(const double* __restrict__ data,
                         const double* __restrict__ data2, BinArgs A,
                         const double* __restrict__ k2edges_g,
                         const double* __restrict__ muedges_g,
                         double* __restrict__ gout)
{
    const int NB = (A.nx_edges + 1) * (A.nmu_edges + 1);
    const int nfields = 3 + 2 * A.nell;
    const int d2 = (int)A.d2;
    const bool comp1_on = A.fuse && A.win1 >= 0;
    const bool comp2_on = A.fuse && A.win2 >= 0;
    const bool same_comp = (A.win2 == A.win1 && A.interl2 == A.interl1);

    extern __shared__ double lh[];
    double* h = gout;
    double* p = lh;
    if (LDS) {
        for (int i = threadIdx.x; i < NB * nfields; i += blockDim.x)
            lh[i] = 0.0;
        h = lh;
        p += NB * nfields;
    }
    double* ke = p;
    p += A.nx_edges;
    double* me = p;
    p += A.nmu_edges;
    double* ctz1 = p;
    p += d2;
    double* ctz2 = p;
    for (int i = threadIdx.x; i < A.nx_edges; i += blockDim.x)
        ke[i] = k2edges_g[i];
    for (int i = threadIdx.x; i < A.nmu_edges; i += blockDim.x)
        me[i] = muedges_g[i];
    for (int i = threadIdx.x; i < d2; i += blockDim.x) {
        const double fz = freq_half(i + A.o2, A.n2);
        const double wz = 2.0 * M_PI * fz / (double)A.n2;
        ctz1[i] = comp1_on
            ? nbk_comp_factor1(A.win1, A.interl1, wz) : 1.0;
        ctz2[i] = (comp2_on && !same_comp)
            ? nbk_comp_factor1(A.win2, A.interl2, wz) : ctz1[i];
    }
    __syncthreads();
    const double k2last = ke[A.nx_edges - 1];

    const int64_t n3[3] = {A.n0, A.n1, A.n2};
    const int64_t nlines = A.d0 * A.d1;
    // NL=32 lines per group: at d2 ~= n2/2+1 (513 for 1024^3) a group
    // holds NL*d2 elements, and with per-thread runs of R the active
    // thread count is NL*d2/R — NL=8/R=8 left half a 1024-thread block
    // IDLE at C4 (4104/8 = 513 workers).  NL=32/R=16 gives 16416/16 =
    // 1026 active threads AND halves the flush rate (measured: R=16
    // -2 ms/step at C4 over R=8; R=32 would idle half the block
    // again).
    constexpr int NL = 32;
    constexpr int R = 16;
    __shared__ double lsxy[NL], ldot[NL], lc1[NL], lc2[NL];
    __shared__ unsigned char lz0[NL];
    const int64_t ngroups = (nlines + NL - 1) / NL;
    for (int64_t grp = blockIdx.x; grp < ngroups; grp += gridDim.x) {
        const int64_t line0 = grp * NL;
        __syncthreads();
        if (threadIdx.x < NL && line0 + threadIdx.x < nlines) {
            const int64_t line = line0 + threadIdx.x;
            const int64_t l0 = line / A.d1;
            const int64_t l1 = line - l0 * A.d1;
            const double f_a0 = freq_axis(A.a0, l0 + A.o0, n3[A.a0], 0);
            const double f_a1 = freq_axis(A.a1, l1 + A.o1, n3[A.a1], 0);
            const double fx = (A.a0 == 0) ? f_a0 : f_a1;
            const double fy = (A.a0 == 0) ? f_a1 : f_a0;
            const double kx = fx * A.k0x;
            const double ky = fy * A.k0y;
            lsxy[threadIdx.x] = kx * kx + ky * ky;
            ldot[threadIdx.x] = kx * A.losx + ky * A.losy;
            lz0[threadIdx.x] = (fx == 0.0 && fy == 0.0);
            if (comp1_on) {
                const double wx = 2.0 * M_PI * fx / (double)A.n0;
                const double wy = 2.0 * M_PI * fy / (double)A.n1;
                lc1[threadIdx.x] = nbk_comp_factor1(A.win1, A.interl1, wx)
                                 * nbk_comp_factor1(A.win1, A.interl1, wy);
            } else {
                lc1[threadIdx.x] = 1.0;
            }
            if (comp2_on && !same_comp) {
                const double wx = 2.0 * M_PI * fx / (double)A.n0;
                const double wy = 2.0 * M_PI * fy / (double)A.n1;
                lc2[threadIdx.x] = nbk_comp_factor1(A.win2, A.interl2, wx)
                                 * nbk_comp_factor1(A.win2, A.interl2, wy);
            } else {
                lc2[threadIdx.x] = lc1[threadIdx.x];
            }
        }
        __syncthreads();
        const int64_t nlive = (nlines - line0 < NL) ? nlines - line0
                                                    : (int64_t)NL;
        const int64_t nelem = nlive * d2;

        int cbin = -1, cbx = -1, cbmu = -1;
        double ak = 0.0, amu = 0.0, aw = 0.0;
        cdouble ay[NBK_MAX_ELL];
        for (int e = 0; e < A.nell; e++) ay[e] = {0.0, 0.0};

        for (int64_t base = (int64_t)threadIdx.x * R; base < nelem;
             base += (int64_t)blockDim.x * R) {
            int sub = (int)(base / d2);
            int l2 = (int)(base - (int64_t)sub * d2);
            const int rmax = (int)((nelem - base < R) ? nelem - base
                                                      : (int64_t)R);
            for (int r = 0; r < rmax; r++, l2++) {
                if (l2 == d2) { l2 = 0; sub++; }
                const double fz = freq_half(l2 + A.o2, A.n2);
                const double kz = fz * A.k0z;
                const double k2 = lsxy[sub] + kz * kz;
                if (k2 >= k2last) continue;
                const double kmag = sqrt(k2);
                double mu = ldot[sub] + kz * A.losz;
                mu = (kmag == 0.0) ? 0.0 : mu / kmag;

                const bool nonsingular = fz > 0.0;
                const double w = nonsingular ? 2.0 : 1.0;

                // warm-start digitize from the previous bin
                int bx;
                if (cbx >= 0
                    && (cbx == 0 || ke[cbx - 1] <= k2)
                    && (cbx == A.nx_edges || k2 < ke[cbx]))
                    bx = cbx;
                else
                    bx = dig(ke, A.nx_edges, k2);
                int bmu;
                if (cbmu >= 0
                    && (cbmu == 0 || me[cbmu - 1] <= mu)
                    && (cbmu == A.nmu_edges || mu < me[cbmu]))
                    bmu = cbmu;
                else
                    bmu = dig(me, A.nmu_edges, mu);
                const int bin = bx * (A.nmu_edges + 1) + bmu;

                const int64_t idx = (line0 + sub) * (int64_t)d2 + l2;
                cdouble v;
                if (A.fuse) {
                    cdouble a = {data[2 * idx], data[2 * idx + 1]};
                    const double fac1 = lc1[sub] * ctz1[l2];
                    if (comp1_on) a = cscale(a, fac1);
                    cdouble b;
                    if (data2 == data && same_comp) {
                        b = a;
                    } else {
                        b = {data2[2 * idx], data2[2 * idx + 1]};
                        if (comp2_on)
                            b = cscale(b, same_comp ? fac1
                                       : lc2[sub] * ctz2[l2]);
                    }
                    cdouble pv = cmul(a, cconj(b));
                    v = {pv.re * A.volume, pv.im * A.volume};
                    if (A.clear_zero && lz0[sub] && fz == 0.0)
                        v = {0.0, 0.0};
                } else {
                    v = {data[2 * idx], data[2 * idx + 1]};
                }

                if (bin != cbin) {
                    if (cbin >= 0)
                        accum_sums<LDS>(h, NB, cbin, ak, amu, aw, ay,
                                        A.nell);
                    cbin = bin; cbx = bx; cbmu = bmu;
                    ak = 0.0; amu = 0.0; aw = 0.0;
                    for (int e = 0; e < A.nell; e++) ay[e] = {0.0, 0.0};
                }
                ak += kmag * w;
                amu += mu * w;
                aw += w;

                double Pm1 = 0.0, P = 1.0;
                int e = 0;
                for (int l = 0; e < A.nell; l++) {
                    if (l > 0) {
                        const double Pn = ((2 * l - 1) * mu * P
                                           - (l - 1) * Pm1) / l;
                        Pm1 = P;
                        P = Pn;
                    }
                    if (l == A.ells[e]) {
                        cdouble wy = cscale(v, P);
                        if (nonsingular) {
                            if (l % 2) wy = {0.0, 2.0 * wy.im};
                            else wy = {2.0 * wy.re, 0.0};
                        }
                        wy = cscale(wy, 2.0 * l + 1.0);
                        ay[e].re += wy.re;
                        ay[e].im += wy.im;
                        e++;
                    }
                }
            }
        }
        if (cbin >= 0)
            accum_sums<LDS>(h, NB, cbin, ak, amu, aw, ay, A.nell);
    }

    if (LDS) {
        __syncthreads();
        for (int i = threadIdx.x; i < NB * nfields; i += blockDim.x)
            if (lh[i] != 0.0) atomicAdd(&gout[i], lh[i]);
    }
}

// shared launcher for the plain (pre-materialized p3d) and fused
// (compensate+power on the fly) binning passes
static int launch_bin(const double* d1, const double* d2, BinArgs& A,
                      const double* kedges, int64_t nx_edges,
                      const double* muedges, int64_t nmu_edges,
                      const int64_t nmesh[3], const double box[3],
                      const int64_t dims[3], const int64_t off[3],
                      const int axis_map[3],
                      const double los[3], const int* ells, int nell,
                      int real_field,
                      double* xsum, double* musum, double* Nsum,
                      double* ysum, void* stream);

}  // namespace

extern "C" int nbk_bin_power_f64(const double* cplx, const int64_t nmesh[3],
                                 const double box[3],
                                 const int64_t dims[3], const int64_t off[3],
                                 const int axis_map[3],
                                 const double* kedges, int64_t nx_edges,
                                 const double* muedges, int64_t nmu_edges,
                                 const double los[3],
                                 const int* ells, int nell,
                                 int real_field,
                                 double* xsum, double* musum, double* Nsum,
                                 double* ysum, void* stream)
{
    BinArgs A;
    A.fuse = 0;
    A.win1 = A.win2 = -1;
    A.interl1 = A.interl2 = 0;
    A.clear_zero = 0;
    A.volume = 1.0;
    return launch_bin(cplx, cplx, A, kedges, nx_edges, muedges, nmu_edges,
                      nmesh, box, dims, off, axis_map, los, ells, nell,
                      real_field, xsum, musum, Nsum, ysum, stream);
}

extern "C" int nbk_power_bin_f64(const double* c1, const double* c2,
                                 double volume,
                                 int window1, int interlaced1,
                                 int window2, int interlaced2,
                                 int clear_zero_mode,
                                 const int64_t nmesh[3], const double box[3],
                                 const int64_t dims[3], const int64_t off[3],
                                 const int axis_map[3],
                                 const double* kedges, int64_t nx_edges,
                                 const double* muedges, int64_t nmu_edges,
                                 const double los[3],
                                 const int* ells, int nell,
                                 double* xsum, double* musum, double* Nsum,
                                 double* ysum, void* stream)
{
    if (window1 > 2 || window2 > 2) {
        NBK_SET_ERR("nbk_power_bin_f64: bad window (%d, %d)",
                    window1, window2);
        return NBK_ERR_ARG;
    }
    BinArgs A;
    A.fuse = 1;
    A.win1 = window1; A.interl1 = interlaced1;
    A.win2 = window2; A.interl2 = interlaced2;
    A.clear_zero = clear_zero_mode;
    A.volume = volume;
    return launch_bin(c1, c2 ? c2 : c1, A, kedges, nx_edges, muedges,
                      nmu_edges, nmesh, box, dims, off, axis_map, los,
                      ells, nell, /*real_field=*/0,
                      xsum, musum, Nsum, ysum, stream);
}

namespace {

static int launch_bin(const double* d1, const double* d2, BinArgs& A,
                      const double* kedges, int64_t nx_edges,
                      const double* muedges, int64_t nmu_edges,
                      const int64_t nmesh[3], const double box[3],
                      const int64_t dims[3], const int64_t off[3],
                      const int axis_map[3],
                      const double los[3], const int* ells, int nell,
                      int real_field,
                      double* xsum, double* musum, double* Nsum,
                      double* ysum, void* stream)
{
    (void)musum; (void)Nsum; (void)ysum;
    if (nell > NBK_MAX_ELL) {
        NBK_SET_ERR("nbk_bin_power_f64: at most %d multipoles", NBK_MAX_ELL);
        return NBK_ERR_ARG;
    }
    // outputs must be one contiguous block [xsum|musum|Nsum|ysum] — the
    // Python layer allocates them together; enforce the layout.
    const int64_t NB = (nx_edges + 1) * (nmu_edges + 1);
    if (musum != xsum + NB || Nsum != xsum + 2 * NB
        || (nell > 0 && ysum != xsum + 3 * NB)) {
        NBK_SET_ERR("nbk_bin_power_f64: outputs must be contiguous "
                    "[xsum|musum|Nsum|ysum]");
        return NBK_ERR_ARG;
    }

    A.n0 = nmesh[0]; A.n1 = nmesh[1]; A.n2 = nmesh[2];
    A.d0 = dims[0]; A.d1 = dims[1]; A.d2 = dims[2];
    A.o0 = off[0]; A.o1 = off[1]; A.o2 = off[2];
    if (axis_map) { A.a0 = axis_map[0]; A.a1 = axis_map[1]; A.a2 = axis_map[2]; }
    else { A.a0 = 0; A.a1 = 1; A.a2 = 2; }
    // coordinate scale per axis: wavenumber 2 pi f / L for the complex
    // field; for the REAL field the Python/oracle coordinate recipe is
    // fl(fl(f * L) / N) — multiply by the box FIRST, divide by N in
    // the kernel — NOT f * fl(L/N): the configuration lattice spacing
    // equals the default r-bin width, so nearly every point sits
    // exactly on a bin edge and the rounding chain must match
    // bit-for-bit (fuzz seeds 51/55/63 caught the f*H form flipping
    // r-bin membership across dozens of lattice shells)
    if (real_field) {
        A.k0x = box[0];
        A.k0y = box[1];
        A.k0z = box[2];
    } else {
        A.k0x = 2.0 * M_PI / box[0];
        A.k0y = 2.0 * M_PI / box[1];
        A.k0z = 2.0 * M_PI / box[2];
    }
    A.real_field = real_field;
    A.losx = los[0]; A.losy = los[1]; A.losz = los[2];
    A.nx_edges = (int)nx_edges;
    A.nmu_edges = (int)nmu_edges;
    A.nell = nell;
    for (int e = 0; e < NBK_MAX_ELL; e++)
        A.ells[e] = e < nell ? ells[e] : -1;

    const int nfields = 3 + 2 * nell;
    const size_t lds_bytes = ((size_t)NB * nfields + nx_edges + nmu_edges)
                             * sizeof(double);
    const int64_t nlines = A.d0 * A.d1;
    const int64_t ngroups = (nlines + 31) / 32;   // NL in kbin_run (the
                                                  // legacy kbin uses 8 —
                                                  // grid is capped far
                                                  // below either)
    int64_t g = ngroups;
    // bounded: each block's LDS histogram flush costs NB*nfields global
    // atomics, and the atomic pipe runs at ~25 G op/s (count_probe) —
    // the cap is tunable for experiments via NBK_BIN_GRID
    static int64_t gcap = 0;
    if (!gcap) {
        const char* e = getenv("NBK_BIN_GRID");
        gcap = e ? atoll(e) : 4096;   // 8192 was best for the old
                                      // per-element kernel; with
                                      // kbin_run's register merging
                                      // fewer blocks (= fewer LDS
                                      // flushes) measured ~1 ms better
                                      // at C4 (r02 tune sweep)
        if (gcap < 256 || gcap > 65536) gcap = 8192;
    }
    if (g > gcap) g = gcap;
    if (g < 1) g = 1;

    hipStream_t s = (hipStream_t)stream;
    // gfx950 has 160 KiB LDS per CU; dynamic allocations above the 64 KiB
    // default need the attribute raised once.  The per-block histogram is
    // worth ~1 occupancy: the global-atomic fallback costs seconds at
    // 1024^3 x Nmu=5 (every element 5+ HBM atomics).
    const bool fast = (A.a2 == 2) && !real_field;
    if (fast) {
        // kbin_run: edges + the two z compensation tables always live
        // in LDS; the histograms do when they fit
        const size_t tab = ((size_t)nx_edges + nmu_edges + 2 * A.d2)
                           * sizeof(double);
        const size_t full = (size_t)NB * nfields * sizeof(double) + tab;
        if (full <= 160 * 1024) {
            static size_t raised = 0;
            if (full > 64 * 1024 && full > raised) {
                (void)hipFuncSetAttribute(
                    reinterpret_cast<const void*>(&kbin_run<true>),
                    hipFuncAttributeMaxDynamicSharedMemorySize,
                    (int)full);
                raised = full;
            }
            hipLaunchKernelGGL(kbin_run<true>, dim3((uint32_t)g),
                               dim3(1024), full, s, d1, d2, A, kedges,
                               muedges, xsum);
        } else {
            static size_t raised_f = 0;
            if (tab > 64 * 1024 && tab > raised_f) {
                (void)hipFuncSetAttribute(
                    reinterpret_cast<const void*>(&kbin_run<false>),
                    hipFuncAttributeMaxDynamicSharedMemorySize,
                    (int)tab);
                raised_f = tab;
            }
            hipLaunchKernelGGL(kbin_run<false>, dim3((uint32_t)g),
                               dim3(1024), tab, s, d1, d2, A, kedges,
                               muedges, xsum);
        }
    } else if (lds_bytes <= 160 * 1024) {
        static size_t raised = 0;
        if (lds_bytes > 64 * 1024 && lds_bytes > raised) {
            (void)hipFuncSetAttribute(
                reinterpret_cast<const void*>(&kbin<true>),
                hipFuncAttributeMaxDynamicSharedMemorySize,
                (int)lds_bytes);
            raised = lds_bytes;
        }
        hipLaunchKernelGGL(kbin<true>, dim3((uint32_t)g), dim3(1024),
                           lds_bytes, s, d1, d2, A, kedges, muedges, xsum);
    } else {
        hipLaunchKernelGGL(kbin<false>, dim3((uint32_t)g), dim3(1024), 0, s,
                           d1, d2, A, kedges, muedges, xsum);
    }
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

}  // namespace
