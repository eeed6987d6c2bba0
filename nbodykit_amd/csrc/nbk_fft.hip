// Hand-written power-of-two FFT passes for the 3D R2C/C2R transform
// (replaces pfft-python / FFTW on the path; pmesh r2c/c2r semantics:
// forward normalized by 1/Ntotal — nbodykit/algorithms/fftpower.py:126-128,
// mockmaker.py:27-36 — and c2r the unnormalized inverse).
//
// Structure: the Python layer composes the 3D transform from
//   nbk_fft_r2c_z     — real -> half-complex along the contiguous axis,
//                       via the packed-real trick (N reals as N/2 complex,
//                       one N/2-point FFT, then the untwiddle split)
//   nbk_fft_c_strided — in-place complex pass along a strided axis
// with the RCCL alltoall pencil transpose between ranks (Python side).
//
// Each line is transformed in LDS by an in-place radix-4 DIT network
// (fused radix-2 pairs): coalesced natural-order global loads are
// scattered bit-reversed into LDS, log4(N) butterfly stages follow, and
// the natural-order result is stored back coalesced.  Twiddles come from
// a cached per-length device table (W_N^j, j = 0..N/2) computed once on
// host; stage twiddles index it as W_len^p = W_N^{p*N/len}.
// Everything is f64 — the path is HBM-bound (~160 B/cell for the 3D
// transform), so VALU cost of f64 butterflies is not the limit.
#include "nbk_common.h"
#include <map>
#include <mutex>

namespace {

// ---- twiddle table cache ---------------------------------------------
std::mutex table_mutex;
std::map<int64_t, double*> twiddle_tables;   // N -> device W_N^[0..N/2]

bool is_pow2(int64_t n) { return n > 0 && (n & (n - 1)) == 0; }

// returns device pointer to W_N table (N/2+1 cdouble), or nullptr on error
double* get_twiddles(int64_t N) {
    std::lock_guard<std::mutex> lock(table_mutex);
    auto it = twiddle_tables.find(N);
    if (it != twiddle_tables.end()) return it->second;

    const int64_t m = N / 2 + 1;
    double* host = (double*)malloc(m * 2 * sizeof(double));
    if (!host) return nullptr;
    for (int64_t j = 0; j < m; j++) {
        const double ang = -2.0 * M_PI * (double)j / (double)N;
        host[2 * j] = cos(ang);
        host[2 * j + 1] = sin(ang);
    }
    double* dev = nullptr;
    if (hipMalloc(&dev, m * 2 * sizeof(double)) != hipSuccess) {
        free(host);
        return nullptr;
    }
    if (hipMemcpy(dev, host, m * 2 * sizeof(double),
                  hipMemcpyHostToDevice) != hipSuccess) {
        free(host);
        (void)hipFree(dev);
        return nullptr;
    }
    free(host);
    twiddle_tables[N] = dev;
    return dev;
}

__device__ __forceinline__ int bitrev(int j, int bits) {
    return (int)(__brev((unsigned)j) >> (32 - bits));
}

// Radix-4 variant: pairs of radix-2 DIT stages fused into one 4-point
// butterfly over the SAME bit-reversed input order, halving the
// __syncthreads count and LDS round trips (the strided passes are
// sync/latency bound — DESIGN.md round-2 lever 1).  For odd log2(m)
// one multiply-free radix-2 stage (W_2^0 = 1) runs first.  Algebra:
// with w = W_{2L}^p, the fused (len=L, len=2L) pair on slots
// (p, p+h, p+2h, p+3h), h = L/2, is the standard radix-4 DIT butterfly
// with x1/x2 exchanged (bit-reversal vs digit-reversal order):
//   b1 = x2 w, b2 = x1 w^2, b3 = x3 w^3
//   e0 = x0 + b2, e1 = x0 - b2, o0 = b1 + b3, o1 = b1 - b3
//   y = (e0+o0, e1 -/+ i*o1, e0-o0, e1 +/- i*o1)   (forward/inverse)
// Validated element-exact against numpy fft for m = 4..2048.
// PITCH is the LDS storage row stride in cdoubles (>= TI; the fused
// x+bin kernel pads to TI+1 so its per-thread j-runs read across LDS
// banks instead of landing on two)
// mstop: run stages while 2*len <= mstop (pass m for the full
// transform; m/2 leaves the FINAL radix-4 stage to the caller — the
// fused x+bin kernel runs it in registers and never writes it back)
template <bool INV>
__device__ void lds_fft4(cdouble* buf, int m, int TI, int PITCH,
                         const cdouble* __restrict__ table,
                         int mstop) {
    const int T = blockDim.x;
    const int tid = threadIdx.x;
    const int bits = 31 - __clz((unsigned)m);
    int len = 2;
    if (bits & 1) {
        for (int w = tid; w < (m >> 1) * TI; w += T) {
            const int c = w % TI;
            const int j = w / TI;
            const int i0 = (2 * j) * PITCH + c;
            const int i1 = i0 + PITCH;
            const cdouble u = buf[i0];
            const cdouble v = buf[i1];
            buf[i0] = cadd(u, v);
            buf[i1] = csub(u, v);
        }
        nbk_sync_lds();
        len = 4;
    }
    for (; 2 * len <= mstop; len <<= 2) {
        const int h = len >> 1;
        const int tw = m / len;              // W_{2L}^p = table[p*tw]
        for (int w = tid; w < (m >> 2) * TI; w += T) {
            const int c = w % TI;
            const int j = w / TI;
            const int grp = j / h;
            const int pos = j - grp * h;
            const int base = (grp * (len << 1) + pos) * PITCH + c;
            const int s = h * PITCH;
            cdouble w1 = table[pos * tw];
            cdouble w2 = table[2 * pos * tw];
            if (INV) { w1.im = -w1.im; w2.im = -w2.im; }
            const cdouble w3 = cmul(w1, w2);
            const cdouble x0 = buf[base];
            const cdouble x1 = buf[base + s];
            const cdouble x2 = buf[base + 2 * s];
            const cdouble x3 = buf[base + 3 * s];
            const cdouble b1 = cmul(x2, w1);
            const cdouble b2 = cmul(x1, w2);
            const cdouble b3 = cmul(x3, w3);
            const cdouble e0 = cadd(x0, b2);
            const cdouble e1 = csub(x0, b2);
            const cdouble o0 = cadd(b1, b3);
            const cdouble o1 = csub(b1, b3);
            buf[base] = cadd(e0, o0);
            buf[base + 2 * s] = csub(e0, o0);
            const cdouble io1 = INV ? cdouble{-o1.im, o1.re}
                                    : cdouble{o1.im, -o1.re};
            buf[base + s] = cadd(e1, io1);
            buf[base + 3 * s] = csub(e1, io1);
        }
        nbk_sync_lds();
    }
}

// ---- z-axis real <-> half-complex ------------------------------------

// one contiguous real line of nz doubles per block -> nz/2+1 complex
__global__ void kfft_r2c_z(const double* __restrict__ real,
                           double* __restrict__ cplx,
                           int64_t nz, double scale,
                           const cdouble* __restrict__ table /* W_nz */)
{
    extern __shared__ cdouble buf[];          // nz/2 entries
    const int m = (int)(nz >> 1);
    const int bits = 31 - __clz((unsigned)m);
    const int64_t line = blockIdx.x;

    const cdouble* g = (const cdouble*)(real + line * nz);  // packed pairs
    for (int q = threadIdx.x; q < m; q += blockDim.x)
        buf[bitrev(q, bits)] = g[q];
    __syncthreads();

    lds_fft4<false>(buf, m, 1, 1, table, m);

    // untwiddle split: X[k] = E[k] + W_nz^k * O[k], k = 0..m
    cdouble* out = (cdouble*)cplx + line * (m + 1);
    for (int k = threadIdx.x; k <= m; k += blockDim.x) {
        const cdouble Zk = buf[k == m ? 0 : k];
        const cdouble Zm = buf[(m - k) % m];
        const cdouble E = cscale(cadd(Zk, cconj(Zm)), 0.5);
        const cdouble D = csub(Zk, cconj(Zm));
        const cdouble O = {0.5 * D.im, -0.5 * D.re};    // D * (-i/2)
        const cdouble X = cadd(E, cmul(table[k], O));
        out[k] = cscale(X, scale);
    }
}

// inverse: nz/2+1 complex -> nz reals, unnormalized
// (so that c2r(r2c(x, scale=1)) == nz * x per line)
__global__ void kfft_c2r_z(const double* __restrict__ cplx,
                           double* __restrict__ real,
                           int64_t nz,
                           const cdouble* __restrict__ table /* W_nz */)
{
    extern __shared__ cdouble smem[];
    const int m = (int)(nz >> 1);
    const int bits = 31 - __clz((unsigned)m);
    cdouble* xin = smem;                      // m+1 entries
    cdouble* buf = smem + (m + 1);            // m entries
    const int64_t line = blockIdx.x;

    const cdouble* g = (const cdouble*)cplx + line * (m + 1);
    for (int k = threadIdx.x; k <= m; k += blockDim.x) {
        cdouble v = g[k];
        // FFTW/numpy c2r convention: the self-conjugate DC and Nyquist
        // bins contribute only their real part (pmesh sits on FFTW; the
        // interlaced combine feeds bins with nonzero imaginary parts
        // there, source/mesh/catalog.py:341-351)
        if (k == 0 || k == m) v.im = 0.0;
        xin[k] = v;
    }
    __syncthreads();

    // rebuild packed spectrum: Z[k] = E[k] + i * O[k],
    // E = (X[k]+conj(X[m-k]))/2, O = conj(W^k) (X[k]-conj(X[m-k]))/2
    for (int k = threadIdx.x; k < m; k += blockDim.x) {
        const cdouble Xk = xin[k];
        const cdouble Xm = cconj(xin[m - k]);
        const cdouble E = cscale(cadd(Xk, Xm), 0.5);
        const cdouble WO = cscale(csub(Xk, Xm), 0.5);
        const cdouble O = cmul(cconj(table[k]), WO);
        const cdouble Z = {E.re - O.im, E.im + O.re};    // E + i O
        buf[bitrev(k, bits)] = Z;
    }
    __syncthreads();

    lds_fft4<true>(buf, m, 1, 1, table, m);

    // unpack: line[2t] = 2 Re(z[t]), line[2t+1] = 2 Im(z[t])
    cdouble* out = (cdouble*)(real + line * nz);
    for (int t = threadIdx.x; t < m; t += blockDim.x)
        out[t] = {2.0 * buf[t].re, 2.0 * buf[t].im};
}

// ---- strided complex pass --------------------------------------------

// one (outer line, TI-column tile) per block; element j of column
// (o, i) at cplx[o*ostride + j*stride + i]
template <bool INV>
__global__ void kfft_c_strided(double* __restrict__ data,
                               int nfft, int64_t stride,
                               int64_t ostride, int64_t n_inner,
                               int TI, int tiles,
                               const cdouble* __restrict__ table /* W_2nfft */)
{
    extern __shared__ cdouble buf[];          // nfft * TI
    const int bits = 31 - __clz((unsigned)nfft);
    const int64_t o = blockIdx.x / tiles;
    const int64_t c0 = (int64_t)(blockIdx.x % tiles) * TI;
    cdouble* g = (cdouble*)data + o * ostride + c0;
    const int ncol = (int)min((int64_t)TI, n_inner - c0);

    // register-staged loads, 8 per thread per round: every global load
    // of a round issues BEFORE the first LDS store, so the whole batch
    // is in flight together.  The pass is latency-bound (each row is
    // one 64 B line megabytes from the next); interleaving load + LDS
    // store per element serializes on the waitcnt and was measured at
    // <3 GB/s per block.
    const int total = nfft * TI;
    for (int base = 0; base < total; base += (int)blockDim.x * 8) {
        cdouble r[8];
        #pragma unroll
        for (int q = 0; q < 8; q++) {
            const int w = base + q * (int)blockDim.x + (int)threadIdx.x;
            if (w < total) {
                const int c = w % TI;
                r[q] = (c < ncol)
                    ? g[(int64_t)(w / TI) * stride + c]
                    : cdouble{0.0, 0.0};
            }
        }
        #pragma unroll
        for (int q = 0; q < 8; q++) {
            const int w = base + q * (int)blockDim.x + (int)threadIdx.x;
            if (w < total)
                buf[bitrev(w / TI, bits) * TI + (w % TI)] = r[q];
        }
    }
    __syncthreads();

    lds_fft4<INV>(buf, nfft, TI, TI, table, nfft);

    for (int w = threadIdx.x; w < nfft * TI; w += blockDim.x) {
        const int c = w % TI;
        const int j = w / TI;
        if (c < ncol)
            g[(int64_t)j * stride + c] = buf[j * TI + c];
    }
}

int check_len(int64_t n, const char* who) {
    if (!is_pow2(n) || n < 8 || n > 4096) {
        NBK_SET_ERR("%s: length %lld unsupported (need power of two in "
                    "[8, 4096])", who, (long long)n);
        return NBK_ERR_UNSUPPORTED;
    }
    return NBK_OK;
}

// ---- fused x-pass FFT + compensate + power + project_to_basis --------
// (auto power, non-interlaced — the FFTPower flagship path): the LAST
// strided pass of the forward 3D FFT feeds each transformed x-line
// straight into the project_to_basis sums, so the finished complex
// field is never written to HBM and the separate binning pass never
// reads it (~17 GB saved per 1024^3 FFTPower), and x-lines wholly
// beyond the last k-edge (ky^2+kz^2 >= k2_last — ~21% of columns at
// the default kmax, the part of the y-z plane outside the inscribed
// circle) are skipped before their loads are even issued.
//
// Bin assignment is bit-identical to kbin_run/kbin (nbk_bin.hip):
//   k2 = (kx*kx + ky*ky) + kz*kz      [fl(fl(kx2+ky2)+kz2)]
//   mu = ((kx lx + ky ly) + kz lz)/|k|
// with the x-dependent terms (kx = fl(fx*k0x), fl(kx*kx),
// fl(kx*losx)) recomputed per element from an n0-entry LDS kx table,
// and the compensation composed as (cx*cy)*cz per-axis products,
// exactly the fast kbin_run composition.  The x-line FFT is the same
// bit-reversed-load + lds_fft4 network as kfft_c_strided, so element
// values match the unfused x-pass bit-for-bit.
#ifndef NBK_MAX_ELL
#define NBK_MAX_ELL 8
#endif

struct XBinArgs {
    int64_t n0, n1, n2;        // global REAL mesh dims
    int64_t nzh;               // n2/2 + 1
    int64_t n_inner;           // local flattened (y, zh) column count
    int64_t y_off;             // global y offset of the local block
    double k0x, k0y, k0z;      // 2 pi / BoxSize
    double losx, losy, losz;
    int nx_edges, nmu_edges;   // edge COUNTS (bins + 1)
    int nell;
    int ells[NBK_MAX_ELL];
    int win1, interl1;         // -1: no compensation
    int clear_zero;
    double volume;
    // uniform-edge digitize hints (the deferred path always has
    // arange/linspace edge grids): bin = guess from (v - lo) * inv_d,
    // then CORRECTED against the exact edge values, so the assignment
    // stays bit-identical to numpy.digitize while the 9-probe binary
    // search chain collapses to ~2 probes
    double k_lo, k_invd;       // on |k| (edges are squared k)
    double mu_lo, mu_invd;
    int TI, tiles;
};

// exact digitize with a uniform-grid starting guess: the guess only
// picks the starting bin; the while-loops compare against the exact
// edge array with numpy.digitize's own predicate
__device__ __forceinline__ int digx_guess(const double* __restrict__ e,
                                          int ne, double x, double guess)
{
    int b = (int)guess + 1;
    if (b < 0) b = 0;
    if (b > ne) b = ne;
    while (b > 0 && x < e[b - 1]) b--;
    while (b < ne && e[b] <= x) b++;
    return b;
}

// numpy.digitize(x, edges): IDENTICAL to nbk_bin.hip `dig` — keep in
// lockstep (bin-edge parity depends on it)
__device__ __forceinline__ int digx(const double* __restrict__ edges,
                                    int nedges, double x) {
    int lo = 0, hi = nedges;
    while (lo < hi) {
        const int mid = (lo + hi) >> 1;
        if (edges[mid] <= x) lo = mid + 1;
        else hi = mid;
    }
    return lo;
}

// PHASES: bit 1 = run the FFT network, bit 2 = run the bin phase
// (3 = the real kernel; other values exist only for the perf
// decomposition tool, NBK_XBIN_PHASES).  MAXE bounds the per-thread
// multipole accumulator (register pressure: nell == 1 — every plain
// 1d/2d run — must not pay for 8).
//
// Software pipeline: tile t+1's strided global loads are issued right
// after its column constants publish, BEFORE tile t's FFT + bin phases
// run, so the HBM latency hides under compute (measured phases at C4:
// loads 2.4 ms + FFT 2.9 ms + bin 4.7 ms, perfectly serial without
// this).  Column constants and live-counts are double-buffered; the
// FFT tile is carried across the compute phases in registers (4
// cdoubles per thread — the launcher caps n0*TI at 4*blockDim).
//
// Auto power only, so the per-element value is real EXACTLY:
// Im(a conj(a)) = fl(-re*im) + fl(im*re) = 0 under -ffp-contract=off.
// The imaginary multipole sums are therefore never accumulated (their
// output planes stay zero, matching the unfused kernel's zeros).
// IL: interlaced-pair mode — a SECOND pre-x field (the half-cell
// shifted paint) rides along; both tiles are FFT'd and combined per
// element as c = a/2 + b/2 exp(i k.H/2) (nbk_interlace_combine's
// formula, phases composed from an LDS x-table times a per-column
// factor) before compensation and binning.  The self-conjugate z
// planes (iz = 0 and, for even n2, the Nyquist plane) are SKIPPED —
// their Hermitian projection couples (-kx,-ky) across columns, so the
// host handles those two planes with the standalone kernels.
template <int PHASES, int MAXE, bool IL>
__global__ void kxfft_bin(const double* __restrict__ data,
                          const double* __restrict__ data2, XBinArgs A,
                          const double* __restrict__ k2edges_g,
                          const double* __restrict__ muedges_g,
                          const cdouble* __restrict__ table /* W_2n0 */,
                          double* __restrict__ gout)
{
    const int NB = (A.nx_edges + 1) * (A.nmu_edges + 1);
    const int nfields = 3 + 2 * A.nell;
    const int n0 = (int)A.n0;
    const int TI = A.TI;
    const int W = TI + 1;          // padded LDS pitch: the bin phase's
                                   // per-thread j-runs stride across
                                   // banks instead of landing on two
    const int T = blockDim.x;
    const int t = threadIdx.x;
    const int bits = 31 - __clz((unsigned)n0);

    // LDS: [fft buf (16B-aligned, first, pitch W) | hist | k2 edges |
    //        mu edges | x-compensation table]  (kx is recomputed per
    //        element — same fl arithmetic as the coordinate tables)
    extern __shared__ cdouble smem[];                  // (shared decl
                                                       // with kfft_c2r_z)
    cdouble* buf = smem;                               // n0 * W
    cdouble* buf2 = IL ? smem + (size_t)n0 * W : nullptr;
    cdouble* pxv = IL ? smem + 2 * (size_t)n0 * W : nullptr;  // n0
    double* h = (double*)smem
        + 2 * (size_t)n0 * W * (IL ? 2 : 1)
        + (IL ? 2 * (size_t)n0 : 0);                   // NB * nfields
    double* ke = h + (size_t)NB * nfields;             // nx_edges
    double* me = ke + A.nx_edges;                      // nmu_edges
    double* cxv = me + A.nmu_edges;                    // n0
    __shared__ double cky2[2][16], ckz2[2][16], ckyl[2][16],
                      ckzl[2][16], ccy[2][16], ccz[2][16], cph[2][32];
    __shared__ unsigned char cw2[2][16], czl[2][16], cskip[2][16];
    __shared__ int s_nlive[2];

    const bool comp_on = A.win1 >= 0;
    for (int i = t; i < NB * nfields; i += T) h[i] = 0.0;
    for (int i = t; i < A.nx_edges; i += T) ke[i] = k2edges_g[i];
    for (int i = t; i < A.nmu_edges; i += T) me[i] = muedges_g[i];
    for (int j = t; j < n0; j += T) {
        const double fx = freq_full(j, A.n0);
        cxv[j] = comp_on
            ? nbk_comp_factor1(A.win1, A.interl1,
                               2.0 * M_PI * fx / (double)A.n0)
            : 1.0;
        if (IL) {
            double sp, cp;
            sincos(M_PI * fx / (double)A.n0, &sp, &cp);
            pxv[j] = {cp, sp};      // exp(i kx Hx / 2)
        }
    }
    // the skip threshold, read straight from global (one scalar; the
    // LDS copy is not published until the first sync)
    const double k2last = k2edges_g[A.nx_edges - 1];


    // ---- pipeline stages (macros so the prologue and loop share the
    //      exact code) --------------------------------------------------
#define XBIN_SETUP(tile_, q_)                                              \
    if (t < 64) {                                                          \
        int live_ = 0;                                                     \
        if (t < TI) {                                                      \
            const int64_t cg_ = (int64_t)(tile_) * TI + t;                 \
            if ((tile_) < A.tiles && cg_ < A.n_inner) {                    \
                const int64_t iyl_ = cg_ / A.nzh;                          \
                const int64_t iz_ = cg_ - iyl_ * A.nzh;                    \
                const double fy_ = freq_full(iyl_ + A.y_off, A.n1);        \
                const double fz_ = freq_half(iz_, A.n2);                   \
                const double ky_ = fy_ * A.k0y;                            \
                const double kz_ = fz_ * A.k0z;                            \
                cky2[q_][t] = ky_ * ky_;                                   \
                ckz2[q_][t] = kz_ * kz_;                                   \
                ckyl[q_][t] = ky_ * A.losy;                                \
                ckzl[q_][t] = kz_ * A.losz;                                \
                cw2[q_][t] = (fz_ > 0.0);                                  \
                czl[q_][t] = (fy_ == 0.0 && fz_ == 0.0);                   \
                if (IL) {                                                  \
                    double sp_, cp_;                                       \
                    sincos(M_PI * (fy_ / (double)A.n1                      \
                                   + fz_ / (double)A.n2), &sp_, &cp_);     \
                    cph[q_][2 * t] = cp_;                                  \
                    cph[q_][2 * t + 1] = sp_;                              \
                }                                                          \
                if (comp_on) {                                             \
                    ccy[q_][t] = nbk_comp_factor1(A.win1, A.interl1,       \
                        2.0 * M_PI * fy_ / (double)A.n1);                  \
                    ccz[q_][t] = nbk_comp_factor1(A.win1, A.interl1,       \
                        2.0 * M_PI * fz_ / (double)A.n2);                  \
                } else {                                                   \
                    ccy[q_][t] = 1.0;                                      \
                    ccz[q_][t] = 1.0;                                      \
                }                                                          \
                /* the line's k2 minimum is at kx = 0 exactly, where   */  \
                /* fl(0 + ky2) = ky2; fl-addition is monotone           */  \
                live_ = (cky2[q_][t] + ckz2[q_][t] < k2last) ? 1 : 0;      \
                /* IL: the self-conjugate planes go to the host path */    \
                if (IL && (iz_ == 0                                        \
                           || (A.n2 % 2 == 0 && iz_ == A.nzh - 1)))        \
                    live_ = 0;                                             \
            }                                                              \
            cskip[q_][t] = (unsigned char)(!live_);                        \
        }                                                                  \
        const unsigned long long m_ = __ballot(live_);                     \
        if (t == 0) s_nlive[q_] = __popcll(m_);                            \
    }

#define XBIN_LOAD(tile_, q_)                                               \
    {                                                                      \
        const cdouble* g_ = (const cdouble*)data + (int64_t)(tile_) * TI;  \
        const cdouble* g2_ = IL                                            \
            ? (const cdouble*)data2 + (int64_t)(tile_) * TI : nullptr;     \
        _Pragma("unroll")                                                  \
        for (int q = 0; q < NLD; q++) {                                    \
            const int w_ = q * T + t;                                      \
            if (w_ < n0 * TI) {                                            \
                const int c_ = w_ % TI;                                    \
                const bool dead_ = cskip[q_][c_];                          \
                const int64_t at_ = (int64_t)(w_ / TI) * A.n_inner + c_;   \
                r[q] = dead_ ? cdouble{0.0, 0.0} : g_[at_];                \
                if (IL)                                                    \
                    r2[q] = dead_ ? cdouble{0.0, 0.0} : g2_[at_];          \
            }                                                              \
        }                                                                  \
    }

#define XBIN_STORE                                                         \
    _Pragma("unroll")                                                      \
    for (int q = 0; q < NLD; q++) {                                        \
        const int w_ = q * T + t;                                          \
        if (w_ < n0 * TI) {                                                \
            const int at_ = bitrev(w_ / TI, bits) * W + (w_ % TI);         \
            buf[at_] = r[q];                                               \
            if (IL)                                                        \
                buf2[at_] = r2[q];                                         \
        }                                                                  \
    }

    // CONTIGUOUS tile chunks per block (not grid-strided): adjacent
    // tiles are adjacent (y, z) columns — L2/TLB locality for the
    // pipelined strided loads
    const int64_t chunk = (A.tiles + gridDim.x - 1) / gridDim.x;
    const int64_t tend = ((int64_t)(blockIdx.x + 1) * chunk < A.tiles)
        ? (int64_t)(blockIdx.x + 1) * chunk : A.tiles;
    // IL carries two tiles in registers — the launcher halves the
    // n0*TI cap so the held-register count stays constant
    constexpr int NLD = IL ? 2 : 4;
    cdouble r[NLD] = {};
    cdouble r2[IL ? NLD : 1] = {};
    int64_t tile = (int64_t)blockIdx.x * chunk;
    int p = 0;

    XBIN_SETUP(tile, 0)
    nbk_sync_lds();             // publishes h/ke/me/cxv and consts[0]
    if (tile < tend)
        XBIN_LOAD(tile, 0)

    for (; tile < tend; tile++, p ^= 1) {
        XBIN_STORE              // regs -> buf (prev bin done: end sync)
        const int64_t nxt = tile + 1;
        XBIN_SETUP(nxt, p ^ 1)
        nbk_sync_lds();         // buf ready; consts[p^1] published
                                // (LDS-only wait: the loads issued for
                                // THIS tile were consumed by the store
                                // above; the next tile's loads, issued
                                // below, must survive every barrier
                                // until the next store)
        if (nxt < tend)
            XBIN_LOAD(nxt, p ^ 1)   // in flight under FFT + bin below

        if (s_nlive[p] > 0) {
            // all stages but the last (for IL, the shifted field runs
            // its FULL transform — its elements are read back per
            // element in the fused last stage below)
            if (PHASES & 1) {
                lds_fft4<false>(buf, n0, TI, W, table, n0 / 2);
                if (IL)
                    lds_fft4<false>(buf2, n0, TI, W, table, n0);
            }

            // FUSED FINAL STAGE + binning: the last radix-4 butterfly's
            // four outputs (j = pos, pos+n0/4, pos+n0/2, pos+3n0/4 of
            // column c = w%TI) are binned straight from REGISTERS —
            // the finished line is never written back to LDS and the
            // bin phase re-reads nothing (the v4 blocked-run readback
            // was 16-way bank-conflicted and dominated the kernel).
            if (PHASES & 2)
            for (int w = t; w < (n0 >> 2) * TI; w += T) {
                const int c = w % TI;
                if (cskip[p][c]) continue;
                const int pos = w / TI;
                const int len = n0 >> 1;     // final stage geometry
                const int s_ = (len >> 1) * W;
                const int base = pos * W + c;
                const int tw = 2;            // m / len
                const cdouble w1 = table[pos * tw];
                const cdouble w2 = table[2 * pos * tw];
                const cdouble w3 = cmul(w1, w2);
                const cdouble x0 = buf[base];
                const cdouble x1 = buf[base + s_];
                const cdouble x2 = buf[base + 2 * s_];
                const cdouble x3 = buf[base + 3 * s_];
                const cdouble b1 = cmul(x2, w1);
                const cdouble b2 = cmul(x1, w2);
                const cdouble b3 = cmul(x3, w3);
                const cdouble e0 = cadd(x0, b2);
                const cdouble e1 = csub(x0, b2);
                const cdouble o0 = cadd(b1, b3);
                const cdouble o1 = csub(b1, b3);
                const cdouble io1 = {o1.im, -o1.re};
                cdouble y4[4];
                y4[0] = cadd(e0, o0);
                y4[1] = cadd(e1, io1);
                y4[2] = csub(e0, o0);
                y4[3] = csub(e1, io1);

                const double ky2 = cky2[p][c];
                const double kz2 = ckz2[p][c];
                const double kyl = ckyl[p][c];
                const double kzl = ckzl[p][c];
                const double ccyv = ccy[p][c];
                const double cczv = ccz[p][c];
                const cdouble phc = IL
                    ? cdouble{cph[p][2 * c], cph[p][2 * c + 1]}
                    : cdouble{1.0, 0.0};
                const bool nonsingular = cw2[p][c];
                const double wgt = nonsingular ? 2.0 : 1.0;
                const bool col_zero = czl[p][c];

                #pragma unroll
                for (int q4 = 0; q4 < 4; q4++) {
                    const int j = pos + q4 * (n0 >> 2);
                    const double fx = freq_full(j, A.n0);
                    const double kx = fx * A.k0x;
                    const double k2 = (kx * kx + ky2) + kz2;
                    if (k2 >= k2last) continue;
                    const double kmag = sqrt(k2);
                    double mu = (kx * A.losx + kyl) + kzl;
                    mu = (kmag == 0.0) ? 0.0 : mu / kmag;
                    const int bx = digx_guess(
                        ke, A.nx_edges, k2,
                        (kmag - A.k_lo) * A.k_invd);
                    const int bmu = digx_guess(
                        me, A.nmu_edges, mu,
                        (mu - A.mu_lo) * A.mu_invd);
                    const int bin = bx * (A.nmu_edges + 1) + bmu;

                    cdouble a = y4[q4];
                    if (IL) {
                        const cdouble b = buf2[j * W + c];
                        const cdouble ph = cmul(pxv[j], phc);
                        a = cadd(cscale(a, 0.5),
                                 cscale(cmul(b, ph), 0.5));
                    }
                    if (comp_on)
                        a = cscale(a, (cxv[j] * ccyv) * cczv);
                    // auto power: Im(a conj(a)) == 0 exactly
                    double vre = (a.re * a.re - a.im * (-a.im))
                                 * A.volume;
                    if (A.clear_zero && col_zero && j == 0)
                        vre = 0.0;

                    atomicAdd(&h[bin], kmag * wgt);
                    atomicAdd(&h[NB + bin], mu * wgt);
                    atomicAdd(&h[2 * NB + bin], wgt);
                    double Pm1 = 0.0, P = 1.0;
                    int e = 0;
                    for (int l = 0; e < A.nell && e < MAXE; l++) {
                        if (l > 0) {
                            const double Pn = ((2 * l - 1) * mu * P
                                               - (l - 1) * Pm1) / l;
                            Pm1 = P;
                            P = Pn;
                        }
                        if (l == A.ells[e]) {
                            double wr = vre * P;
                            if (nonsingular)
                                wr = (l % 2) ? 0.0 : 2.0 * wr;
                            atomicAdd(&h[(3 + 2 * e) * NB + bin],
                                      wr * (2.0 * l + 1.0));
                            e++;
                        }
                    }
                }
            }
        }
        nbk_sync_lds();         // bin done before the next tile's store
    }
#undef XBIN_SETUP
#undef XBIN_LOAD
#undef XBIN_STORE
    nbk_sync_lds();
    for (int i = t; i < NB * nfields; i += T)
        if (h[i] != 0.0) atomicAdd(&gout[i], h[i]);
}

}  // namespace

extern "C" int nbk_fft_r2c_z(const double* real, double* cplx,
                             int64_t nlines, int64_t nz, double scale,
                             void* stream)
{
    int rc = check_len(nz, "nbk_fft_r2c_z");
    if (rc) return rc;
    if (nlines == 0) return NBK_OK;
    double* table = get_twiddles(nz);
    if (!table) { NBK_SET_ERR("twiddle alloc failed"); return NBK_ERR_HIP; }
    const int block = (int)std::min<int64_t>(512, nz / 4);
    const size_t shmem = (size_t)(nz / 2) * sizeof(cdouble);
    hipLaunchKernelGGL(kfft_r2c_z, dim3((uint32_t)nlines), dim3(block), shmem,
                       (hipStream_t)stream, real, cplx, nz, scale,
                       (const cdouble*)table);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_fft_c2r_z(const double* cplx, double* real,
                             int64_t nlines, int64_t nz, void* stream)
{
    int rc = check_len(nz, "nbk_fft_c2r_z");
    if (rc) return rc;
    if (nlines == 0) return NBK_OK;
    double* table = get_twiddles(nz);
    if (!table) { NBK_SET_ERR("twiddle alloc failed"); return NBK_ERR_HIP; }
    const int block = (int)std::min<int64_t>(512, nz / 4);
    const size_t shmem = (size_t)(nz + 1) * sizeof(cdouble);
    hipLaunchKernelGGL(kfft_c2r_z, dim3((uint32_t)nlines), dim3(block), shmem,
                       (hipStream_t)stream, cplx, real, nz,
                       (const cdouble*)table);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_fft_c_strided(double* cplx, int64_t nfft, int64_t stride,
                                 int64_t n_outer, int64_t outer_stride,
                                 int64_t n_inner, int sign, void* stream)
{
    int rc = check_len(nfft, "nbk_fft_c_strided");
    if (rc) return rc;
    if (n_outer == 0 || n_inner == 0) return NBK_OK;
    // W_{2*nfft} table so stage twiddles use even indices
    double* table = get_twiddles(2 * nfft);
    if (!table) { NBK_SET_ERR("twiddle alloc failed"); return NBK_ERR_HIP; }

    // inner-tile width: 4 columns x 16 B = 64 B contiguous per element
    // row; LDS = nfft*TI*16 B <= 64 KiB keeps 2 blocks (8 waves) per CU.
    // (TI=8 with 128 KiB LDS was tried under radix-2: full 128 B lines
    // but 1 block/CU — measured 16.2 vs 13.0 ms per 1024^3 pass;
    // latency hiding won.  NBK_FFT_TI re-opens the experiment for the
    // radix-4 network.)
    static int TI0 = 0;
    if (!TI0) {
        const char* e = getenv("NBK_FFT_TI");
        TI0 = e ? atoi(e) : 4;
        if (TI0 < 1 || TI0 > 16) TI0 = 4;
    }
    int TI = TI0;
    // clamp to the 64 KiB two-blocks-per-CU budget unless the env
    // explicitly asked for a bigger tile (gfx950 LDS tops at 160 KiB)
    const int64_t lds_cap = (TI0 != 4) ? 160 * 1024 : 65536;
    while ((int64_t)nfft * TI * (int64_t)sizeof(cdouble) > lds_cap && TI > 1)
        TI >>= 1;
    if (TI > n_inner) TI = (int)n_inner;
    const int tiles = (int)((n_inner + TI - 1) / TI);
    const int64_t grid = n_outer * tiles;
    if (grid > 0x7fffffff) {
        NBK_SET_ERR("nbk_fft_c_strided: grid too large");
        return NBK_ERR_ARG;
    }
    // block size: more waves per CU = more outstanding strided lines
    // (the pass is latency-bound, ~1.3 TB/s at 256 threads); overridable
    // for experiments via NBK_FFT_BLOCK
    static int block = 0;
    if (!block) {
        const char* e = getenv("NBK_FFT_BLOCK");
        block = e ? atoi(e) : 1024;
        if (block < 64 || block > 1024) block = 1024;
    }
    const size_t shmem = (size_t)nfft * TI * sizeof(cdouble);
    if (shmem > 65536) {
        static size_t raised_fwd = 0, raised_inv = 0;
        size_t& raised = (sign < 0) ? raised_fwd : raised_inv;
        if (shmem > raised) {
            (void)hipFuncSetAttribute(
                sign < 0
                    ? reinterpret_cast<const void*>(&kfft_c_strided<false>)
                    : reinterpret_cast<const void*>(&kfft_c_strided<true>),
                hipFuncAttributeMaxDynamicSharedMemorySize, (int)shmem);
            raised = shmem;
        }
    }

    if (sign < 0)
        hipLaunchKernelGGL(kfft_c_strided<false>, dim3((uint32_t)grid),
                           dim3(block), shmem, (hipStream_t)stream,
                           cplx, (int)nfft, stride, outer_stride, n_inner,
                           TI, tiles, (const cdouble*)table);
    else
        hipLaunchKernelGGL(kfft_c_strided<true>, dim3((uint32_t)grid),
                           dim3(block), shmem, (hipStream_t)stream,
                           cplx, (int)nfft, stride, outer_stride, n_inner,
                           TI, tiles, (const cdouble*)table);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_fft_x_bin_f64(const double* data, const double* data2,
                                 const int64_t nmesh[3],
                                 int64_t n_inner, int64_t y_off,
                                 const double box[3],
                                 int window1, int interlaced1,
                                 int clear_zero, double volume,
                                 const double* k2edges, int64_t nx_edges,
                                 const double* muedges, int64_t nmu_edges,
                                 const double dig_hints[4],
                                 const double los[3],
                                 const int* ells, int nell,
                                 double* out_sums, void* stream)
{
    const bool il = data2 != nullptr;
    int rc = check_len(nmesh[0], "nbk_fft_x_bin_f64");
    if (rc) return rc;
    if (n_inner == 0) return NBK_OK;
    if (nell > NBK_MAX_ELL) {
        NBK_SET_ERR("nbk_fft_x_bin_f64: at most %d multipoles",
                    NBK_MAX_ELL);
        return NBK_ERR_ARG;
    }
    if (window1 > 2) {
        NBK_SET_ERR("nbk_fft_x_bin_f64: bad window %d", window1);
        return NBK_ERR_ARG;
    }
    double* table = get_twiddles(2 * nmesh[0]);
    if (!table) { NBK_SET_ERR("twiddle alloc failed"); return NBK_ERR_HIP; }

    XBinArgs A;
    A.n0 = nmesh[0]; A.n1 = nmesh[1]; A.n2 = nmesh[2];
    A.nzh = nmesh[2] / 2 + 1;
    A.n_inner = n_inner;
    A.y_off = y_off;
    A.k0x = 2.0 * M_PI / box[0];
    A.k0y = 2.0 * M_PI / box[1];
    A.k0z = 2.0 * M_PI / box[2];
    A.losx = los[0]; A.losy = los[1]; A.losz = los[2];
    A.nx_edges = (int)nx_edges;
    A.nmu_edges = (int)nmu_edges;
    A.nell = nell;
    for (int e = 0; e < NBK_MAX_ELL; e++)
        A.ells[e] = e < nell ? ells[e] : -1;
    A.win1 = window1;
    A.interl1 = interlaced1;
    A.clear_zero = clear_zero;
    A.volume = volume;
    A.k_lo = dig_hints[0];
    A.k_invd = dig_hints[1];
    A.mu_lo = dig_hints[2];
    A.mu_invd = dig_hints[3];

    const int64_t NB = (nx_edges + 1) * (nmu_edges + 1);
    const int nfields = 3 + 2 * nell;
    const size_t fixed = ((size_t)NB * nfields + nx_edges + nmu_edges
                          + nmesh[0]) * sizeof(double);
    static int TI0 = 0;
    if (!TI0) {
        const char* e = getenv("NBK_XBIN_TI");
        TI0 = e ? atoi(e) : 4;
        if (TI0 < 1 || TI0 > 16) TI0 = 4;
    }
    int TI = TI0;
    // the pipeline carries a tile in 4 registers per thread (2 + 2
    // when an interlaced pair rides along)
    const int64_t regcap = il ? 2 * 1024 : 4 * 1024;
    while (TI > 1 && (int64_t)nmesh[0] * TI > regcap)
        TI >>= 1;
    // buf pitch is TI+1 (bank-spread padding for the bin phase's
    // runs); interlaced mode holds two tiles + the x phase table
    const size_t fixed_il = fixed
        + (il ? (size_t)nmesh[0] * sizeof(cdouble) : 0);
    const int nbufs = il ? 2 : 1;
    while (TI > 1 && fixed_il + (size_t)nbufs * nmesh[0] * (TI + 1)
                         * sizeof(cdouble) > 160 * 1024)
        TI >>= 1;
    if ((int64_t)nmesh[0] * TI > regcap) {
        NBK_SET_ERR("nbk_fft_x_bin_f64: n0 too large for the register "
                    "pipeline — use the unfused path");
        return NBK_ERR_UNSUPPORTED;
    }
    const size_t shmem = fixed_il
        + (size_t)nbufs * nmesh[0] * (TI + 1) * sizeof(cdouble);
    if (shmem > 160 * 1024) {
        NBK_SET_ERR("nbk_fft_x_bin_f64: LDS budget exceeded "
                    "(%zu B) — use the unfused path", shmem);
        return NBK_ERR_UNSUPPORTED;
    }
    A.TI = TI;
    A.tiles = (int)((n_inner + TI - 1) / TI);

    // persistent grid: each block flushes one LDS histogram at the end
    // (the global atomic pipe runs ~25 G op/s — cap the flush count)
    static int64_t gcap = 0;
    if (!gcap) {
        const char* e = getenv("NBK_XBIN_GRID");
        gcap = e ? atoll(e) : 2048;
        if (gcap < 256 || gcap > 65536) gcap = 2048;
    }
    int64_t g = A.tiles;
    if (g > gcap) g = gcap;

    // NBK_XBIN_PHASES: perf-decomposition only (1 = loads+FFT,
    // 2 = loads+bin, 0 = loads only); the product always runs 3
    static int phases = 0;
    if (!phases) {
        const char* e = getenv("NBK_XBIN_PHASES");
        phases = e ? atoi(e) : 3;
        if (phases < 0 || phases > 3) phases = 3;
        phases |= 4;            // mark initialized
    }
    // MAXE sizes the per-thread multipole registers: <.,1,.> serves
    // every plain 1d/2d run, <.,8,.> any poles request; the perf
    // decomposition phases exist only at MAXE = 8, non-interlaced
    const int ph = phases & 3;
    int variant;               // 0..3: <3,1,F> <3,8,F> <3,1,T> <3,8,T>
    if (ph != 3)
        variant = 4 + ph;      // 4..6: phases 0..2 at <ph,8,F>
    else
        variant = (il ? 2 : 0) + (nell <= 1 ? 0 : 1);
    const void* fns[7] = {
        reinterpret_cast<const void*>(&kxfft_bin<3, 1, false>),
        reinterpret_cast<const void*>(&kxfft_bin<3, 8, false>),
        reinterpret_cast<const void*>(&kxfft_bin<3, 1, true>),
        reinterpret_cast<const void*>(&kxfft_bin<3, 8, true>),
        reinterpret_cast<const void*>(&kxfft_bin<0, 8, false>),
        reinterpret_cast<const void*>(&kxfft_bin<1, 8, false>),
        reinterpret_cast<const void*>(&kxfft_bin<2, 8, false>),
    };
    static size_t raised[7] = {};
    if (shmem > 64 * 1024 && shmem > raised[variant]) {
        (void)hipFuncSetAttribute(
            fns[variant], hipFuncAttributeMaxDynamicSharedMemorySize,
            (int)shmem);
        raised[variant] = shmem;
    }
    switch (variant) {
    case 0:
        hipLaunchKernelGGL((kxfft_bin<3, 1, false>), dim3((uint32_t)g),
                           dim3(1024), shmem, (hipStream_t)stream, data,
                           data2, A, k2edges, muedges,
                           (const cdouble*)table, out_sums);
        break;
    case 1:
        hipLaunchKernelGGL((kxfft_bin<3, 8, false>), dim3((uint32_t)g),
                           dim3(1024), shmem, (hipStream_t)stream, data,
                           data2, A, k2edges, muedges,
                           (const cdouble*)table, out_sums);
        break;
    case 2:
        hipLaunchKernelGGL((kxfft_bin<3, 1, true>), dim3((uint32_t)g),
                           dim3(1024), shmem, (hipStream_t)stream, data,
                           data2, A, k2edges, muedges,
                           (const cdouble*)table, out_sums);
        break;
    case 3:
        hipLaunchKernelGGL((kxfft_bin<3, 8, true>), dim3((uint32_t)g),
                           dim3(1024), shmem, (hipStream_t)stream, data,
                           data2, A, k2edges, muedges,
                           (const cdouble*)table, out_sums);
        break;
    case 4:
        hipLaunchKernelGGL((kxfft_bin<0, 8, false>), dim3((uint32_t)g),
                           dim3(1024), shmem, (hipStream_t)stream, data,
                           data2, A, k2edges, muedges,
                           (const cdouble*)table, out_sums);
        break;
    case 5:
        hipLaunchKernelGGL((kxfft_bin<1, 8, false>), dim3((uint32_t)g),
                           dim3(1024), shmem, (hipStream_t)stream, data,
                           data2, A, k2edges, muedges,
                           (const cdouble*)table, out_sums);
        break;
    default:
        hipLaunchKernelGGL((kxfft_bin<2, 8, false>), dim3((uint32_t)g),
                           dim3(1024), shmem, (hipStream_t)stream, data,
                           data2, A, k2edges, muedges,
                           (const cdouble*)table, out_sums);
        break;
    }
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

// cross-TU handle for the fused paint+z-FFT kernel (nbk_paint.hip)
double* nbk_internal_twiddles(int64_t N) { return get_twiddles(N); }
