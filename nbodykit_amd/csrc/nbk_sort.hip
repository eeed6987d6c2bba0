// Particle locality sorts for the paint kernels.
//
// Two pipelines:
// - nbk_xsort_count/scatter + nbk_bucket_fine (big meshes): a
//   two-level ATOMIC-FREE counting sort.  The global atomic pipe runs
//   at ~25 G op/s regardless of locality (csrc/count_probe.hip), so no
//   stage may use global atomics: the coarse sort by
//   (ix, iy-group) goes through per-chunk count MATRICES (host-scanned,
//   no shared ticket counters) + LDS cursors, and the fine pass runs
//   count + block scan + placement entirely in LDS, one block per
//   coarse bucket.  Row-only fine mode serves the ownership-gather
//   paint (which needs row grouping, not full cell order) and emits
//   the row table it reads.
// - nbk_bucket_count/scatter (small inputs): the single-level counting
//   sort by full cell id with int32 countdown tickets, paired with the
//   wave-merged scatter paint.
//
// Both are non-stable within a bucket/cell (fine: deposits commute)
// and emit SoA (x[n] y[n] z[n]) — the layout the paint kernels read.
#include "nbk_common.h"

namespace {

__device__ __forceinline__ int64_t bucket_of(double x, double y, double z,
                                             double invH0, double invH1,
                                             double invH2, int64_t n0,
                                             int64_t n1, int64_t n2,
                                             int shift) {
    const int64_t ix = wrap_idx((int64_t)floor(x * invH0), n0);
    const int64_t iy = wrap_idx((int64_t)floor(y * invH1), n1);
    const int64_t iz = wrap_idx((int64_t)floor(z * invH2), n2);
    return ((ix * n1 + iy) * n2 + iz) >> shift;
}

// also detects (heuristically, lane-adjacent pairs) whether the input
// is already cell-ordered: scrambled -> *scrambled_flag = 1.  The caller
// skips the scatter pass for ordered input.
__global__ void kbucket_count(const double* __restrict__ pos, int64_t n,
                              int64_t n0, int64_t n1, int64_t n2,
                              double invH0, double invH1, double invH2,
                              int shift,
                              int* __restrict__ counts,
                              int* __restrict__ scrambled_flag)
{
    const int lane = threadIdx.x & 63;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int out_of_order = 0;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < n; i += stride) {
        const int64_t b = bucket_of(pos[3 * i], pos[3 * i + 1],
                                    pos[3 * i + 2], invH0, invH1, invH2,
                                    n0, n1, n2, shift);
        atomicAdd(&counts[b], 1);
        const int64_t b_up = __shfl_up((long long)b, 1, 64);
        if (lane > 0 && b_up > b) out_of_order = 1;
    }
    if (out_of_order) atomicOr(scrambled_flag, 1);
}

// SOA=1: write x/y/z planes (the paint layout); SOA=0: AoS rows (the
// intermediate layout of the two-level sort's coarse pass)
template <int SOA>
__global__ void kbucket_scatter(const double* __restrict__ pos,
                                const double* __restrict__ mass, int64_t n,
                                int64_t n0, int64_t n1, int64_t n2,
                                double invH0, double invH1, double invH2,
                                int shift,
                                int* __restrict__ offsets,
                                double* __restrict__ ox,
                                double* __restrict__ oy,
                                double* __restrict__ oz,
                                double* __restrict__ om)
{
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < n; i += stride) {
        const double x = pos[3 * i], y = pos[3 * i + 1], z = pos[3 * i + 2];
        const int64_t b = bucket_of(x, y, z, invH0, invH1, invH2,
                                    n0, n1, n2, shift);
        // offsets holds the INCLUSIVE bucket cumsum; tickets count DOWN
        // to the exclusive base (saves the host-side shift/int64 cast —
        // within-bucket order is arbitrary either way)
        const int64_t t = (int64_t)(atomicSub(&offsets[b], 1) - 1);
        if (SOA) {
            ox[t] = x;
            oy[t] = y;
            oz[t] = z;
        } else {
            ox[3 * t] = x;
            ox[3 * t + 1] = y;
            ox[3 * t + 2] = z;
        }
        if (mass) om[t] = mass[i];
    }
}

// ---- chunked x-plane pre-sort (the coarse level) ----------------------
// Deterministic counting sort by wrapped x-plane index, with NO global
// atomics: pass A writes one histogram row per chunk to a [nchunks x n0]
// matrix, the host scans it bucket-major, and pass C places each chunk's
// particles from LDS cursors seeded by the scanned bases.  Running the
// fine cell sort on this output makes its scattered counter/ticket
// atomics hit 64-byte lines ~16x each (one x-plane's counters span
// n1*n2*4 B) instead of thrashing the whole multi-GB array.

// coarse key: ix * (n1 >> ys) + (iy >> ys) — ys chosen by the host so
// that BOTH the coarse histogram (nbuckets ints) and the fine pass's
// per-bucket cell window ((1 << ys) * n2 ints) fit in LDS
__device__ __forceinline__ int64_t coarse_key(double x, double y,
                                              double invH0, double invH1,
                                              int64_t n0, int64_t n1,
                                              int ys) {
    const int64_t ix = wrap_idx((int64_t)floor(x * invH0), n0);
    const int64_t iy = wrap_idx((int64_t)floor(y * invH1), n1);
    return ix * (n1 >> ys) + (iy >> ys);
}

__global__ void kxsort_count(const double* __restrict__ pos, int64_t n,
                             int chunk, int64_t n0, int64_t n1, int64_t n2,
                             double invH0, double invH1, double invH2,
                             int ys, int64_t nbuck,
                             int* __restrict__ mat,
                             int* __restrict__ scrambled_flag)
{
    extern __shared__ int hist[];   // nbuck ints
    for (int64_t b = threadIdx.x; b < nbuck; b += blockDim.x) hist[b] = 0;
    __syncthreads();
    const int64_t beg = (int64_t)blockIdx.x * chunk;
    const int64_t end = (beg + chunk < n) ? beg + chunk : n;
    const int lane = threadIdx.x & 63;
    int out_of_order = 0;
    for (int64_t i = beg + threadIdx.x; i < end; i += blockDim.x) {
        const double x = pos[3 * i], y = pos[3 * i + 1],
                     z = pos[3 * i + 2];
        atomicAdd(&hist[coarse_key(x, y, invH0, invH1, n0, n1, ys)], 1);
        if (scrambled_flag) {
            const int64_t b = bucket_of(x, y, z, invH0, invH1, invH2,
                                        n0, n1, n2, 0);
            const int64_t b_up = __shfl_up((long long)b, 1, 64);
            if (lane > 0 && b_up > b) out_of_order = 1;
        }
    }
    if (out_of_order) atomicOr(scrambled_flag, 1);
    __syncthreads();
    for (int64_t b = threadIdx.x; b < nbuck; b += blockDim.x)
        mat[(int64_t)blockIdx.x * nbuck + b] = hist[b];
}

// emits SoA planes (ox/oy/oz) so the fine pass's counting loop can read
// the 8 B/particle y plane instead of 24 B AoS rows — 16 GB less
// traffic at C4 (fine reads y twice: count + scatter)
__global__ void kxsort_scatter(const double* __restrict__ pos,
                               const double* __restrict__ mass, int64_t n,
                               int chunk, int64_t n0, int64_t n1,
                               double invH0, double invH1,
                               int ys, int64_t nbuck,
                               const int* __restrict__ bases,
                               double* __restrict__ ox,
                               double* __restrict__ oy,
                               double* __restrict__ oz,
                               double* __restrict__ om)
{
    extern __shared__ int cur[];    // nbuck running cursors
    for (int64_t b = threadIdx.x; b < nbuck; b += blockDim.x)
        cur[b] = bases[(int64_t)blockIdx.x * nbuck + b];
    __syncthreads();
    const int64_t beg = (int64_t)blockIdx.x * chunk;
    const int64_t end = (beg + chunk < n) ? beg + chunk : n;
    for (int64_t i = beg + threadIdx.x; i < end; i += blockDim.x) {
        const double x = pos[3 * i], y = pos[3 * i + 1],
                     z = pos[3 * i + 2];
        const int64_t k = coarse_key(x, y, invH0, invH1, n0, n1, ys);
        const int64_t t = (int64_t)atomicAdd(&cur[k], 1);
        ox[t] = x;
        oy[t] = y;
        oz[t] = z;
        if (mass) om[t] = mass[i];
    }
}

// ---- count-matrix scan (replaces the torch transpose/cumsum glue) ----
// mat is [nblocks x nbuck] chunk-major.  Produces bases[c][b] =
// (sum of all buckets < b over every chunk) + (sum of bucket b over
// chunks < c) — the cursor seeds kxsort_scatter reads — and
// bucket_bases[0..nbuck] = exclusive bucket totals (the fine pass's
// block ranges).  One coalesced read for the column sums, one
// read+write for the bases: ~1.5 passes over mat vs the ~8 of the
// transpose+cumsum+sub+transpose torch chain it replaces.

// segmented column sums: with only nbuck (~32K) threads the serial
// per-bucket chunk walks ran at 2 waves/CU and cost ~3 ms/step at C4;
// splitting each column into NSEG segments gives NSEG x the threads
// (full occupancy) at ~1.5 extra matrix passes of traffic.
#define NBK_SCAN_SEG 8

__global__ void kscan_partial(const int* __restrict__ mat,
                              int64_t nblocks, int64_t nbuck,
                              int64_t cs /* chunks per segment */,
                              int* __restrict__ partial /* NSEG*nbuck */)
{
    const int64_t flat = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
    if (flat >= nbuck * NBK_SCAN_SEG) return;
    const int64_t s = flat / nbuck;
    const int64_t b = flat - s * nbuck;
    const int64_t c0 = s * cs;
    const int64_t c1 = (c0 + cs < nblocks) ? c0 + cs : nblocks;
    int acc = 0;
    for (int64_t c = c0; c < c1; c++)
        acc += mat[c * nbuck + b];
    partial[s * nbuck + b] = acc;
}

__global__ void kscan_fold(const int* __restrict__ partial,
                           int64_t nbuck, int* __restrict__ colsum)
{
    const int64_t b = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
    if (b >= nbuck) return;
    int s = 0;
    #pragma unroll
    for (int j = 0; j < NBK_SCAN_SEG; j++)
        s += partial[(int64_t)j * nbuck + b];
    colsum[b] = s;
}

// single-block exclusive scan of colsum (nbuck <= 40960): writes the
// exclusive base back into colsum and the inclusive edges into
// bucket_bases[1..nbuck] (bucket_bases[0] = 0)
__global__ void kscan_exclusive(int* __restrict__ colsum, int64_t nbuck,
                                int* __restrict__ bucket_bases)
{
    __shared__ int wsum[17];
    const int T = blockDim.x;
    const int t = threadIdx.x;
    const int S = (int)((nbuck + T - 1) / T);
    int acc = 0;
    for (int j = 0; j < S; j++) {
        const int64_t idx = (int64_t)t * S + j;
        if (idx < nbuck) acc += colsum[idx];
    }
    const int lane = t & 63;
    const int wave = t >> 6;
    int v = acc;
    #pragma unroll
    for (int d = 1; d < 64; d <<= 1) {
        const int u = __shfl_up(v, d, 64);
        if (lane >= d) v += u;
    }
    if (lane == 63) wsum[wave] = v;
    __syncthreads();
    if (t == 0) {
        int run = 0;
        for (int w = 0; w < (T >> 6); w++) {
            const int x = wsum[w];
            wsum[w] = run;
            run += x;
        }
    }
    __syncthreads();
    int run = wsum[wave] + (v - acc);
    for (int j = 0; j < S; j++) {
        const int64_t idx = (int64_t)t * S + j;
        if (idx < nbuck) {
            const int c = colsum[idx];
            colsum[idx] = run;
            run += c;
            bucket_bases[idx + 1] = run;
        }
    }
    if (t == 0) bucket_bases[0] = 0;
}

__global__ void kscan_bases(const int* __restrict__ mat, int64_t nblocks,
                            int64_t nbuck, int64_t cs,
                            const int* __restrict__ partial,
                            const int* __restrict__ excl,
                            int* __restrict__ bases)
{
    const int64_t flat = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
    if (flat >= nbuck * NBK_SCAN_SEG) return;
    const int64_t s = flat / nbuck;
    const int64_t b = flat - s * nbuck;
    int run = excl[b];
    for (int64_t j = 0; j < s; j++)
        run += partial[j * nbuck + b];
    const int64_t c0 = s * cs;
    const int64_t c1 = (c0 + cs < nblocks) ? c0 + cs : nblocks;
    for (int64_t c = c0; c < c1; c++) {
        bases[c * nbuck + b] = run;
        run += mat[c * nbuck + b];
    }
}

// fused fine pass: one block per coarse bucket.  The bucket's particles
// are contiguous (coarse sort) and its cells span a window of
// (1 << ys) * n2 entries, so count + exclusive scan + placement all run
// in LDS — ZERO global atomics.  (The count probe measured the global
// atomic pipe at ~25 G ops/s regardless of locality, which bounded the
// old single-level sort; LDS histograms run at the read bandwidth.)
// Output order = exact cell order, identical to the single-level sort.
__global__ void kbucket_fine(const double* __restrict__ px,
                             const double* __restrict__ py,
                             const double* __restrict__ pz,
                             const double* __restrict__ mass,
                             int64_t n1, int64_t n2,
                             double invH1, double invH2,
                             int ys,
                             const int* __restrict__ bbase, /* nbuck+1 */
                             double* __restrict__ ox,
                             double* __restrict__ oy,
                             double* __restrict__ oz,
                             double* __restrict__ om,
                             int* __restrict__ rowtab /* n0*n1, or NULL:
                                 start index of each (ix, iy) row in the
                                 sorted output — consumed by the gather
                                 paint kernel */)
{
    extern __shared__ int lds[];            // win ints (counts->cursors)
    __shared__ int ssum[16];
    const int64_t win = ((int64_t)1 << ys) * n2;
    const int64_t ymask = ((int64_t)1 << ys) - 1;
    const int T = blockDim.x;
    const int t = threadIdx.x;

    for (int64_t w = t; w < win; w += T) lds[w] = 0;
    __syncthreads();

    const int64_t beg = bbase[blockIdx.x];
    const int64_t end = bbase[blockIdx.x + 1];

    for (int64_t i = beg + t; i < end; i += T) {
        const int64_t iy = wrap_idx((int64_t)floor(py[i] * invH1), n1);
        const int64_t iz = wrap_idx((int64_t)floor(pz[i] * invH2), n2);
        atomicAdd(&lds[(iy & ymask) * n2 + iz], 1);
    }
    __syncthreads();

    // block-wide exclusive scan of lds[0..win), seeded with beg:
    // stripes per thread, wave shfl scan of the stripe sums, serial
    // scan of the wave totals
    const int S = (int)(win / T);           // win % T == 0 (host guard)
    int acc = 0;
    for (int j = 0; j < S; j++) acc += lds[t * S + j];
    const int lane = t & 63;
    const int wave = t >> 6;
    int v = acc;
    #pragma unroll
    for (int d = 1; d < 64; d <<= 1) {
        const int u = __shfl_up(v, d, 64);
        if (lane >= d) v += u;
    }
    if (lane == 63) ssum[wave] = v;
    __syncthreads();
    if (t == 0) {
        int run = 0;
        for (int w = 0; w < (T >> 6); w++) {
            const int x = ssum[w];
            ssum[w] = run;
            run += x;
        }
    }
    __syncthreads();
    int run = (int)beg + ssum[wave] + (v - acc);
    for (int j = 0; j < S; j++) {
        const int c = lds[t * S + j];
        lds[t * S + j] = run;
        run += c;
    }
    __syncthreads();

    if (rowtab) {
        // row r of this bucket starts where its first cell's cursor
        // points (post-scan, pre-scatter)
        const int64_t nyb = n1 >> ys;
        const int64_t ix = blockIdx.x / nyb;
        const int64_t yb = blockIdx.x % nyb;
        for (int r = t; r < (1 << ys); r += T)
            rowtab[ix * n1 + yb * ((int64_t)1 << ys) + r] =
                lds[(int64_t)r * n2];
        __syncthreads();
    }

    for (int64_t i = beg + t; i < end; i += T) {
        const double x = px[i], y = py[i], z = pz[i];
        const int64_t iy = wrap_idx((int64_t)floor(y * invH1), n1);
        const int64_t iz = wrap_idx((int64_t)floor(z * invH2), n2);
        const int slot = atomicAdd(&lds[(iy & ymask) * n2 + iz], 1);
        ox[slot] = x;
        oy[slot] = y;
        oz[slot] = z;
        if (mass) om[slot] = mass[i];
    }
}

// row-only fine pass: when the ownership-gather paint consumes the
// output, full cell order is unnecessary — the gather reads particles
// through ROW intervals (rowtab) and z order within a row is
// irrelevant (unsorted z even reduces its LDS same-address conflicts).
// The window shrinks from (1<<ys)*n2 cells to (1<<ys) rows, so this
// kernel runs at full occupancy with a trivial serial scan.
__global__ void kbucket_fine_rows(const double* __restrict__ px,
                                  const double* __restrict__ py,
                                  const double* __restrict__ pz,
                                  const double* __restrict__ mass,
                                  int64_t n1,
                                  double invH1,
                                  int ys,
                                  const int* __restrict__ bbase,
                                  double* __restrict__ ox,
                                  double* __restrict__ oy,
                                  double* __restrict__ oz,
                                  double* __restrict__ om,
                                  int* __restrict__ rowtab)
{
    extern __shared__ int cur[];            // (1 << ys) cursors
    const int nr = 1 << ys;
    const int64_t ymask = nr - 1;
    const int T = blockDim.x;
    const int t = threadIdx.x;

    for (int r = t; r < nr; r += T) cur[r] = 0;
    __syncthreads();

    const int64_t beg = bbase[blockIdx.x];
    const int64_t end = bbase[blockIdx.x + 1];

    for (int64_t i = beg + t; i < end; i += T) {
        const int64_t iy = wrap_idx((int64_t)floor(py[i] * invH1), n1);
        atomicAdd(&cur[iy & ymask], 1);
    }
    __syncthreads();

    if (t == 0) {
        int run = (int)beg;
        for (int r = 0; r < nr; r++) {
            const int c = cur[r];
            cur[r] = run;
            run += c;
        }
    }
    __syncthreads();

    if (rowtab) {
        const int64_t nyb = n1 >> ys;
        const int64_t ix = blockIdx.x / nyb;
        const int64_t yb = blockIdx.x % nyb;
        for (int r = t; r < nr; r += T)
            rowtab[ix * n1 + yb * (int64_t)nr + r] = cur[r];
        __syncthreads();
    }

    for (int64_t i = beg + t; i < end; i += T) {
        const double x = px[i], y = py[i], z = pz[i];
        const int64_t iy = wrap_idx((int64_t)floor(y * invH1), n1);
        const int slot = atomicAdd(&cur[iy & ymask], 1);
        ox[slot] = x;
        oy[slot] = y;
        oz[slot] = z;
        if (mass) om[slot] = mass[i];
    }
}


// ---- PAIR-BUCKET sort (the fine-pass eliminator) ---------------------
// Key = (x-plane PAIR, y row-GROUP) with nbuck = (n0/2) * (n1 >> ys)
// <= 40960 (i32 LDS histograms), and particles DUPLICATED into every
// y-group their deposit stencil touches (rows [iy+dlo, iy+dhi]; the
// stencil spans <= 5 rows << the group size, so <= 2 copies).  The
// ownership-gather paint then reads each (pair, group) bucket range
// directly — its deposit masks drop the out-of-tile copies — and the
// whole per-row fine pass (56 GB of moves at C4) disappears.  The cost
// moves into the paint's source reads: a pair bucket holds ~3 planes
// of particles, read by each of the pair's 2 tiles.
__device__ __forceinline__ void pair_keys(double x, double y,
                                          double invH0, double invH1,
                                          int64_t n0, int64_t n1, int ys,
                                          int dlo, int dhi,
                                          int64_t* k1, int64_t* k2) {
    const int64_t ix = wrap_idx((int64_t)floor(x * invH0), n0);
    const int64_t iy = (int64_t)floor(y * invH1);
    const int64_t ng = n1 >> ys;
    const int64_t g1 = wrap_idx(iy + dlo, n1) >> ys;
    const int64_t g2 = wrap_idx(iy + dhi, n1) >> ys;
    const int64_t base = (ix >> 1) * ng;
    *k1 = base + g1;
    *k2 = (g2 == g1) ? -1 : base + g2;
}

__global__ void kpsort_count(const double* __restrict__ pos, int64_t n,
                             int chunk, int64_t n0, int64_t n1,
                             double invH0, double invH1,
                             int ys, int dlo, int dhi, int64_t nbuck,
                             int* __restrict__ mat)
{
    extern __shared__ int hist[];   // nbuck ints
    for (int64_t b = threadIdx.x; b < nbuck; b += blockDim.x) hist[b] = 0;
    __syncthreads();
    const int64_t beg = (int64_t)blockIdx.x * chunk;
    const int64_t end = (beg + chunk < n) ? beg + chunk : n;
    for (int64_t i = beg + threadIdx.x; i < end; i += blockDim.x) {
        int64_t k1, k2;
        pair_keys(pos[3 * i], pos[3 * i + 1], invH0, invH1, n0, n1,
                  ys, dlo, dhi, &k1, &k2);
        atomicAdd(&hist[k1], 1);
        if (k2 >= 0) atomicAdd(&hist[k2], 1);
    }
    __syncthreads();
    for (int64_t b = threadIdx.x; b < nbuck; b += blockDim.x)
        mat[(int64_t)blockIdx.x * nbuck + b] = hist[b];
}

__global__ void kpsort_scatter(const double* __restrict__ pos,
                               const double* __restrict__ mass, int64_t n,
                               int chunk, int64_t n0, int64_t n1,
                               double invH0, double invH1,
                               int ys, int dlo, int dhi, int64_t nbuck,
                               const int* __restrict__ bases,
                               int64_t n_out,
                               double* __restrict__ ox,
                               double* __restrict__ oy,
                               double* __restrict__ oz,
                               double* __restrict__ om)
{
    extern __shared__ int cur[];    // nbuck running cursors
    for (int64_t b = threadIdx.x; b < nbuck; b += blockDim.x)
        cur[b] = bases[(int64_t)blockIdx.x * nbuck + b];
    __syncthreads();
    const int64_t beg = (int64_t)blockIdx.x * chunk;
    const int64_t end = (beg + chunk < n) ? beg + chunk : n;
    for (int64_t i = beg + threadIdx.x; i < end; i += blockDim.x) {
        const double x = pos[3 * i], y = pos[3 * i + 1],
                     z = pos[3 * i + 2];
        int64_t k1, k2;
        pair_keys(x, y, invH0, invH1, n0, n1, ys, dlo, dhi, &k1, &k2);
        const double m = mass ? mass[i] : 0.0;
        int64_t t = (int64_t)atomicAdd(&cur[k1], 1);
        ox[t] = x; oy[t] = y; oz[t] = z;
        if (mass) om[t] = m;
        if (k2 >= 0) {
            t = (int64_t)atomicAdd(&cur[k2], 1);
            ox[t] = x; oy[t] = y; oz[t] = z;
            if (mass) om[t] = m;
        }
    }
    (void)n_out;
}

int sgrid(int64_t n) {
    int64_t g = (n + 255) / 256;
    if (g > 1048576) g = 1048576;
    if (g < 1) g = 1;
    return (int)g;
}

}  // namespace

extern "C" int nbk_bucket_count_f64(const double* pos_aos, int64_t n,
                                    const int64_t nmesh[3],
                                    const double box[3], int shift,
                                    int* counts, int* scrambled_flag,
                                    void* stream)
{
    if (n == 0) return NBK_OK;
    hipLaunchKernelGGL(kbucket_count, dim3(sgrid(n)), dim3(256), 0,
                       (hipStream_t)stream, pos_aos, n,
                       nmesh[0], nmesh[1], nmesh[2],
                       nmesh[0] / box[0], nmesh[1] / box[1],
                       nmesh[2] / box[2], shift, counts, scrambled_flag);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_bucket_scatter_f64(const double* pos_aos,
                                      const double* mass, int64_t n,
                                      const int64_t nmesh[3],
                                      const double box[3], int shift,
                                      int soa_out,
                                      int* offsets,
                                      double* pos_out,
                                      double* mass_out, void* stream)
{
    if (n == 0) return NBK_OK;
    hipStream_t s = (hipStream_t)stream;
    if (soa_out)
        hipLaunchKernelGGL(kbucket_scatter<1>, dim3(sgrid(n)), dim3(256),
                           0, s, pos_aos, mass, n,
                           nmesh[0], nmesh[1], nmesh[2],
                           nmesh[0] / box[0], nmesh[1] / box[1],
                           nmesh[2] / box[2], shift, offsets,
                           pos_out, pos_out + n, pos_out + 2 * n,
                           mass_out);
    else
        hipLaunchKernelGGL(kbucket_scatter<0>, dim3(sgrid(n)), dim3(256),
                           0, s, pos_aos, mass, n,
                           nmesh[0], nmesh[1], nmesh[2],
                           nmesh[0] / box[0], nmesh[1] / box[1],
                           nmesh[2] / box[2], shift, offsets,
                           pos_out, nullptr, nullptr, mass_out);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

namespace {
// LDS ceiling for the sort kernels, in int entries (160 KiB on gfx950)
const int64_t NBK_SORT_LDS_INTS = 40960;

void raise_lds(const void* fn, size_t bytes) {
    if (bytes > 64 * 1024)
        (void)hipFuncSetAttribute(fn,
            hipFuncAttributeMaxDynamicSharedMemorySize, (int)bytes);
}
}  // namespace

extern "C" int nbk_xsort_count_f64(const double* pos_aos, int64_t n,
                                   int chunk, const int64_t nmesh[3],
                                   const double box[3], int ys, int* mat,
                                   int* scrambled_flag, void* stream)
{
    if (n == 0) return NBK_OK;
    const int64_t nbuck = nmesh[0] * (nmesh[1] >> ys);
    if (nbuck > NBK_SORT_LDS_INTS || (nmesh[1] % ((int64_t)1 << ys))) {
        NBK_SET_ERR("nbk_xsort_count_f64: bad ys=%d for mesh", ys);
        return NBK_ERR_ARG;
    }
    const int64_t nblocks = (n + chunk - 1) / chunk;
    const size_t lds = (size_t)nbuck * sizeof(int);
    raise_lds(reinterpret_cast<const void*>(&kxsort_count), lds);
    hipLaunchKernelGGL(kxsort_count, dim3((uint32_t)nblocks), dim3(1024),
                       lds, (hipStream_t)stream, pos_aos, n, chunk,
                       nmesh[0], nmesh[1], nmesh[2],
                       nmesh[0] / box[0], nmesh[1] / box[1],
                       nmesh[2] / box[2], ys, nbuck, mat, scrambled_flag);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_xsort_scatter_f64(const double* pos_aos,
                                     const double* mass, int64_t n,
                                     int chunk, const int64_t nmesh[3],
                                     const double box[3], int ys,
                                     const int* bases,
                                     double* pos_out, double* mass_out,
                                     void* stream)
{
    if (n == 0) return NBK_OK;
    const int64_t nbuck = nmesh[0] * (nmesh[1] >> ys);
    if (nbuck > NBK_SORT_LDS_INTS || (nmesh[1] % ((int64_t)1 << ys))) {
        NBK_SET_ERR("nbk_xsort_scatter_f64: bad ys=%d for mesh", ys);
        return NBK_ERR_ARG;
    }
    const int64_t nblocks = (n + chunk - 1) / chunk;
    const size_t lds = (size_t)nbuck * sizeof(int);
    raise_lds(reinterpret_cast<const void*>(&kxsort_scatter), lds);
    hipLaunchKernelGGL(kxsort_scatter, dim3((uint32_t)nblocks),
                       dim3(1024), lds, (hipStream_t)stream, pos_aos,
                       mass, n, chunk, nmesh[0], nmesh[1],
                       nmesh[0] / box[0], nmesh[1] / box[1], ys, nbuck,
                       bases, pos_out, pos_out + n, pos_out + 2 * n,
                       mass_out);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

/* colsum_tmp must hold (NBK_SCAN_SEG + 1) * nbuck ints: segment
 * partials first, the column-sum/exclusive-base vector last. */
extern "C" int nbk_scan_matrix_i32(const int* mat, int64_t nblocks,
                                   int64_t nbuck, int* colsum_tmp,
                                   int* bases, int* bucket_bases,
                                   void* stream)
{
    if (nblocks == 0 || nbuck == 0) return NBK_OK;
    hipStream_t s = (hipStream_t)stream;
    const int T = 256;
    int* partial = colsum_tmp;
    int* colsum = colsum_tmp + NBK_SCAN_SEG * nbuck;
    const int64_t cs = (nblocks + NBK_SCAN_SEG - 1) / NBK_SCAN_SEG;
    const uint32_t gs_ = (uint32_t)((nbuck * NBK_SCAN_SEG + T - 1) / T);
    const uint32_t g = (uint32_t)((nbuck + T - 1) / T);
    hipLaunchKernelGGL(kscan_partial, dim3(gs_), dim3(T), 0, s,
                       mat, nblocks, nbuck, cs, partial);
    hipLaunchKernelGGL(kscan_fold, dim3(g), dim3(T), 0, s,
                       partial, nbuck, colsum);
    hipLaunchKernelGGL(kscan_exclusive, dim3(1), dim3(1024), 0, s,
                       colsum, nbuck, bucket_bases);
    hipLaunchKernelGGL(kscan_bases, dim3(gs_), dim3(T), 0, s,
                       mat, nblocks, nbuck, cs, partial, colsum, bases);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_bucket_fine_f64(const double* pos_soa,
                                   const double* mass, int64_t n,
                                   const int64_t nmesh[3],
                                   const double box[3], int ys,
                                   const int* bucket_bases,
                                   double* soa_out, double* mass_out,
                                   int* rowtab, int rows_only,
                                   void* stream)
{
    if (n == 0) return NBK_OK;
    const int64_t nbuck = nmesh[0] * (nmesh[1] >> ys);
    if (rows_only) {
        if (nmesh[1] % ((int64_t)1 << ys)) {
            NBK_SET_ERR("nbk_bucket_fine_f64: bad ys=%d", ys);
            return NBK_ERR_ARG;
        }
        const size_t lds = ((size_t)1 << ys) * sizeof(int);
        hipLaunchKernelGGL(kbucket_fine_rows, dim3((uint32_t)nbuck),
                           dim3(1024), lds, (hipStream_t)stream,
                           pos_soa, pos_soa + n, pos_soa + 2 * n,
                           mass, nmesh[1], nmesh[1] / box[1], ys,
                           bucket_bases, soa_out, soa_out + n,
                           soa_out + 2 * n, mass_out, rowtab);
        NBK_CHECK_HIP(hipGetLastError());
        return NBK_OK;
    }
    const int64_t win = ((int64_t)1 << ys) * nmesh[2];
    if (win > NBK_SORT_LDS_INTS || (win % 1024)
        || (nmesh[1] % ((int64_t)1 << ys))) {
        NBK_SET_ERR("nbk_bucket_fine_f64: bad ys=%d for mesh", ys);
        return NBK_ERR_ARG;
    }
    const size_t lds = (size_t)win * sizeof(int);
    raise_lds(reinterpret_cast<const void*>(&kbucket_fine), lds);
    hipLaunchKernelGGL(kbucket_fine, dim3((uint32_t)nbuck), dim3(1024),
                       lds, (hipStream_t)stream,
                       pos_soa, pos_soa + n, pos_soa + 2 * n, mass,
                       nmesh[1], nmesh[2],
                       nmesh[1] / box[1], nmesh[2] / box[2], ys,
                       bucket_bases, soa_out, soa_out + n,
                       soa_out + 2 * n, mass_out, rowtab);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_psort_count_f64(const double* pos_aos, int64_t n,
                                   int chunk, const int64_t nmesh[3],
                                   const double box[3], int ys,
                                   int dlo, int dhi, int* mat,
                                   void* stream)
{
    if (n == 0) return NBK_OK;
    const int64_t nbuck = (nmesh[0] >> 1) * (nmesh[1] >> ys);
    if (nbuck > NBK_SORT_LDS_INTS || (nmesh[0] & 1)
        || (nmesh[1] % ((int64_t)1 << ys))
        || (dhi - dlo) >= ((int64_t)1 << ys)) {
        NBK_SET_ERR("nbk_psort_count_f64: bad geometry ys=%d", ys);
        return NBK_ERR_ARG;
    }
    const int64_t nblocks = (n + chunk - 1) / chunk;
    const size_t lds = (size_t)nbuck * sizeof(int);
    raise_lds(reinterpret_cast<const void*>(&kpsort_count), lds);
    hipLaunchKernelGGL(kpsort_count, dim3((uint32_t)nblocks), dim3(1024),
                       lds, (hipStream_t)stream, pos_aos, n, chunk,
                       nmesh[0], nmesh[1],
                       nmesh[0] / box[0], nmesh[1] / box[1],
                       ys, dlo, dhi, nbuck, mat);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_psort_scatter_f64(const double* pos_aos,
                                     const double* mass, int64_t n,
                                     int chunk, const int64_t nmesh[3],
                                     const double box[3], int ys,
                                     int dlo, int dhi,
                                     const int* bases, int64_t n_out,
                                     double* pos_out, double* mass_out,
                                     void* stream)
{
    if (n == 0) return NBK_OK;
    const int64_t nbuck = (nmesh[0] >> 1) * (nmesh[1] >> ys);
    if (nbuck > NBK_SORT_LDS_INTS) {
        NBK_SET_ERR("nbk_psort_scatter_f64: bad ys=%d", ys);
        return NBK_ERR_ARG;
    }
    const int64_t nblocks = (n + chunk - 1) / chunk;
    const size_t lds = (size_t)nbuck * sizeof(int);
    raise_lds(reinterpret_cast<const void*>(&kpsort_scatter), lds);
    hipLaunchKernelGGL(kpsort_scatter, dim3((uint32_t)nblocks),
                       dim3(1024), lds, (hipStream_t)stream, pos_aos,
                       mass, n, chunk, nmesh[0], nmesh[1],
                       nmesh[0] / box[0], nmesh[1] / box[1],
                       ys, dlo, dhi, nbuck, bases, n_out,
                       pos_out, pos_out + n_out, pos_out + 2 * n_out,
                       mass_out);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}
