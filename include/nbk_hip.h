/*
 * nbk_hip.h — C ABI of libnbk_hip.so, the hand-written HIP (gfx950)
 * compute core of nbodykit_amd.
 *
 * This is the internal drop-in boundary defined in SURVEY.md §8(b): the
 * reference (bccp/nbodykit) has no C plugin interface — its operator API
 * is the MeshSource/CatalogSource Python duck type — so this ABI carries
 * exactly the pmesh/pfft operations the Python layer replaces:
 *
 *   nbk_paint_f64        <- pmesh ParticleMesh.paint (CIC/TSC/PCS scatter;
 *                           called at nbodykit/source/mesh/catalog.py:287,
 *                           295-296)
 *   nbk_fft_*            <- pfft-python distributed R2C/C2R
 *                           (pmesh RealField.r2c/ComplexField.c2r; called
 *                           via nbodykit/base/mesh.py:301-304), forward
 *                           normalized by 1/Ntotal
 *   nbk_compensate_f64   <- the Fourier-space window compensations
 *                           (nbodykit/source/mesh/catalog.py:419-594)
 *   nbk_interlace_combine_f64
 *                        <- the interlaced-mesh combine
 *                           c = c1/2 + c2/2 exp(i k.H/2)
 *                           (nbodykit/source/mesh/catalog.py:341-347)
 *   nbk_power3d_f64      <- p3d = c1 conj(c2) * V, zero mode cleared
 *                           (nbodykit/algorithms/fftpower.py:114-128)
 *   nbk_bin_power_f64    <- project_to_basis binning sums
 *                           (nbodykit/algorithms/fftpower.py:507-701)
 *   nbk_readout_nnb_f64 / elementwise helpers — support ops.
 *
 * Conventions:
 *   - plain pointers + sizes only; all device pointers are HIP device
 *     memory owned by the caller (torch allocations in practice, but no
 *     torch types cross this boundary);
 *   - `stream` is a hipStream_t passed as void* (0 = default stream);
 *     all launches are asynchronous on that stream;
 *   - return value: 0 on success, a negative NBK_ERR_* code otherwise;
 *     nbk_last_error_string() describes the most recent failure;
 *   - complex double arrays are interleaved (re,im) — C `double _Complex`
 *     layout, i.e. numpy complex128 / torch complex128;
 *   - the mesh is slab-partitioned along axis 0; every kernel taking
 *     `x0/nx_local` operates on the local slab of a global
 *     (n0, n1, n2) mesh.  Single-GPU: x0 = 0, nx_local = n0.
 */
#ifndef NBK_HIP_H
#define NBK_HIP_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* error codes */
#define NBK_OK            0
#define NBK_ERR_HIP      -1   /* HIP runtime error, see last_error_string */
#define NBK_ERR_ARG      -2   /* bad argument (size, window id, ...) */
#define NBK_ERR_UNSUPPORTED -3 /* e.g. non-power-of-two FFT length */

/* window ids (resampler names in the Python layer) */
#define NBK_WINDOW_CIC 0   /* support 2 */
#define NBK_WINDOW_TSC 1   /* support 3 */
#define NBK_WINDOW_PCS 2   /* support 4 */

/* library info ------------------------------------------------------- */
const char* nbk_version(void);
const char* nbk_last_error_string(void);
/* number of HIP devices visible (also a cheap "runtime works" probe) */
int nbk_device_count(void);

/* paint --------------------------------------------------------------
 * Scatter n particles into the local mesh slab with window `window` and
 * half-cell interlacing shift `shift` (0.0 or 0.5, in mesh units;
 * pm.affine.shift(0.5) at source/mesh/catalog.py:292).
 *
 *   pos     : device, SoA layout, 3 contiguous blocks of n doubles
 *             (x[n], y[n], z[n]) — pos + 0, pos + n, pos + 2n
 *   mass    : device, n doubles, or NULL for unit mass
 *   nmesh   : global mesh size (n0, n1, n2)
 *   box     : BoxSize per axis
 *   x0, nx_local : local slab [x0, x0+nx_local) of axis 0; deposits
 *             whose wrapped global x index falls outside are skipped
 *             (ghost copies on the neighbour rank own them)
 *   mesh    : device, local slab, nx_local*n1*n2 doubles, accumulated
 *             into (hold=True semantics)
 */
int nbk_paint_f64(const double* pos, const double* mass, int64_t n,
                  const int64_t nmesh[3], const double box[3],
                  int window, double shift,
                  double* mesh, int64_t x0, int64_t nx_local,
                  void* stream);

/* paint for cell-sorted input: TSC/PCS deposits accumulate in an LDS
 * window per contiguous particle run and flush once (CIC falls through
 * to nbk_paint_f64).  Same semantics as nbk_paint_f64; the caller must
 * know the input is cell-ordered (the bucket sort's output is). */
int nbk_paint_sorted_f64(const double* pos, const double* mass, int64_t n,
                         const int64_t nmesh[3], const double box[3],
                         int window, double shift,
                         double* mesh, int64_t x0, int64_t nx_local,
                         void* stream);

/* ownership-gather paint for cell-sorted input: requires the row table
 * produced by nbk_bucket_fine_f64 (rowtab[n0*n1+1]: start of each
 * (ix, iy) row in the sorted SoA arrays, sentinel n at the end).  Each
 * block owns an exclusive LDS mesh tile (P x-planes x RG y-rows x n2;
 * the tile shape is picked per window span — single-plane max-RG for
 * CIC, balanced multi-plane for TSC/PCS — within the 160 KiB LDS
 * budget) and gathers from the source rows whose stencils reach it —
 * deposits
 * are LDS f64 adds and the flush is plain stores, NO global atomics
 * (the global atomic pipe is the ~25 G op/s bound on the scatter
 * kernels above).  accumulate=0 overwrites the slab (fresh mesh, saves
 * the read), accumulate=1 adds (hold semantics for chunked paints).
 * Returns NBK_ERR_UNSUPPORTED when no LDS tile fits (fall back to
 * nbk_paint_f64). */
/* pair_gs >= 0: `rowtab` is the PAIR-BUCKET table of the duplicating
 * sort (nbk_psort_*): [(n0/2)*(n1>>pair_gs)+1] exclusive bases, one
 * contiguous range per (x-plane pair, y-group); the tile geometry is
 * pinned to (1 plane x 1<<pair_gs rows) and the deposit masks drop the
 * duplicated out-of-tile copies.  pair_gs = -1: per-row table from
 * nbk_bucket_fine_f64. */
int nbk_paint_gather_f64(const double* pos, const double* mass, int64_t n,
                         const int64_t nmesh[3], const double box[3],
                         int window, double shift, const int* rowtab,
                         double* mesh, int64_t x0, int64_t nx_local,
                         int accumulate, int pair_gs, void* stream);

/* gather paint with the forward z-axis FFT fused into the tile flush:
 * writes the z half-spectrum ((nx_local, n1, n2/2+1) interleaved c128,
 * every element scaled by `scale`) directly — the real mesh never
 * exists in HBM.  Same source/rowtab semantics as nbk_paint_gather_f64;
 * overwrite-only (single paint per output).  n2 must be a power of two
 * in [8, 4096].  The FFT math is identical to nbk_fft_r2c_z. */
int nbk_paint_gather_fft_f64(const double* pos, const double* mass,
                             int64_t n, const int64_t nmesh[3],
                             const double box[3],
                             int window, double shift, const int* rowtab,
                             double* zspec, int64_t x0, int64_t nx_local,
                             double scale, int pair_gs, void* stream);

/* paint locality sort --------------------------------------------------
 * Two-pass counting sort of particles by coarse mesh cell
 * (bucket = wrapped ix * n1 + iy): count, then (after the caller turns
 * counts into exclusive offsets) scatter into SoA output.  Replaces a
 * general radix sort in the paint driver: the deposit kernel wants
 * bucket-local order, not a total order.  Bucket = cell >> shift
 * (shift 0 = per-cell).  `pos_aos` is the (n,3) row-major input;
 * `offsets` (ncells >> shift, int32) holds the INCLUSIVE bucket cumsum
 * and is consumed (tickets count DOWN to the exclusive base) by the
 * scatter; soa_out selects x/y/z planes (paint layout) vs AoS rows.
 * mass may be NULL.  Requires n < 2^31 (int32 tickets).
 */
int nbk_bucket_count_f64(const double* pos_aos, int64_t n,
                         const int64_t nmesh[3], const double box[3],
                         int shift,
                         int* counts, int* scrambled_flag, void* stream);
int nbk_bucket_scatter_f64(const double* pos_aos, const double* mass,
                           int64_t n, const int64_t nmesh[3],
                           const double box[3], int shift, int soa_out,
                           int* offsets,
                           double* pos_out, double* mass_out,
                           void* stream);

/* Two-level atomic-free cell sort (the paint locality pass for big
 * meshes).  The GPU's global-atomic pipe runs at ~25 G ops/s regardless
 * of locality (csrc/count_probe.hip), so the single-level
 * count/scatter above is atomic-bound at 1e9 particles; this pipeline
 * keeps every histogram/cursor in LDS:
 *
 *  coarse key = ix * (n1>>ys) + (iy>>ys), ys chosen so that BOTH
 *  nbuckets = n0*(n1>>ys) and the fine window (1<<ys)*n2 are <= 40960
 *  ints (160 KiB LDS):
 *   - nbk_xsort_count_f64 (pass A): one nbuckets-wide histogram row per
 *     chunk of `chunk` particles into `mat` (ceil(n/chunk) x nbuckets,
 *     int32 row-major), plus the cell-order detection into
 *     scrambled_flag (may be NULL);
 *   - nbk_scan_matrix_i32: device scan of `mat` -> per-chunk cursor
 *     seeds `bases` (chunk-major) and bucket edges `bucket_bases`
 *     (nbuckets+1, exclusive; [nbuckets] = n).  colsum_tmp is nbuckets
 *     ints of scratch;
 *   - nbk_xsort_scatter_f64 (pass C): re-reads each chunk, places SoA
 *     planes (x[n] y[n] z[n] into pos_out) from LDS cursors seeded
 *     with `bases` (deterministic — no global atomics);
 *   - nbk_bucket_fine_f64: one block per coarse bucket over the SoA
 *     coarse output; counts the bucket's cells in an LDS window,
 *     block-scans it, and emits the exact cell-sorted SoA output
 *     (x[n] y[n] z[n]) — no global atomics.
 * Requires power-friendly dims (n1 divisible by 1<<ys, window
 * divisible by 1024) and n < 2^31. */
int nbk_xsort_count_f64(const double* pos_aos, int64_t n, int chunk,
                        const int64_t nmesh[3], const double box[3],
                        int ys, int* mat, int* scrambled_flag,
                        void* stream);
/* PAIR-BUCKET duplicating sort (see nbk_paint_gather_f64 pair_gs):
 * counting sort by (x-plane pair, y row-group) with each particle
 * duplicated into every y-group rows [iy+dlo, iy+dhi] of its deposit
 * stencil touch (<= 2 copies; dhi-dlo must be < 1<<ys).  Same
 * count-matrix/scan/scatter structure as nbk_xsort_*; n_out =
 * bucket_bases[nbuck] after nbk_scan_matrix_i32. */
int nbk_psort_count_f64(const double* pos_aos, int64_t n, int chunk,
                        const int64_t nmesh[3], const double box[3],
                        int ys, int dlo, int dhi, int* mat, void* stream);
int nbk_psort_scatter_f64(const double* pos_aos, const double* mass,
                          int64_t n, int chunk, const int64_t nmesh[3],
                          const double box[3], int ys, int dlo, int dhi,
                          const int* bases, int64_t n_out,
                          double* pos_out, double* mass_out,
                          void* stream);

int nbk_scan_matrix_i32(const int* mat, int64_t nblocks, int64_t nbuck,
                        int* colsum_tmp, int* bases, int* bucket_bases,
                        void* stream);
int nbk_xsort_scatter_f64(const double* pos_aos, const double* mass,
                          int64_t n, int chunk, const int64_t nmesh[3],
                          const double box[3], int ys, const int* bases,
                          double* pos_out, double* mass_out,
                          void* stream);
int nbk_bucket_fine_f64(const double* pos_soa, const double* mass,
                        int64_t n, const int64_t nmesh[3],
                        const double box[3], int ys,
                        const int* bucket_bases,
                        double* soa_out, double* mass_out,
                        int* rowtab, int rows_only, void* stream);

/* readout (gather dual of paint; window 0/1/2 = cic/tsc/pcs, 3 = nnb).
 * Serves FFTRecon's displacement solve (fftrecon.py:246-249) and the
 * LogNormal generator's nnb reads (mockmaker.py:317-319).  Cells outside
 * the local slab contribute 0 — ghost owners add their partials via the
 * host-side exchange. */
int nbk_readout_f64(const double* pos, int64_t n,
                    const int64_t nmesh[3], const double box[3],
                    int window,
                    const double* mesh, int64_t x0, int64_t nx_local,
                    double* out, void* stream);

/* FFTRecon displacement solve (fftrecon.py:222-238):
 * out = i k_axis/k^2 exp(-k^2 R^2/2) / (bias (1 + (f/bias) mu^2)) in */
int nbk_recon_displacement_f64(double* out, const double* in,
                               const int64_t nmesh[3], const double box[3],
                               const int64_t dims[3], const int64_t off[3],
                               int axis, double R, double bias, double f,
                               const double los[3], void* stream);

/* FFT ----------------------------------------------------------------
 * Power-of-two lengths only (8 <= N <= 4096).  The 3D transform is
 * composed by the Python layer from these batched passes (plus the RCCL
 * alltoall pencil transpose between ranks):
 *
 *   z pass  : real <-> half-complex along the contiguous last axis
 *   strided : in-place complex pass along a strided axis (y or x)
 *
 * Layouts: nbk_fft_r2c_z treats `real` as nlines contiguous lines of nz
 * doubles, writing nlines * (nz/2+1) complex outputs; `scale` multiplies
 * every output (the Python layer passes 1/(n0*n1*n2) here so the forward
 * 3D transform matches pmesh's normalization).  nbk_fft_c2r_z is the
 * unnormalized inverse.  nbk_fft_c_strided transforms n_lines lines of
 * length nfft; element j of line (o, i) lives at
 * cplx[o*outer_stride + j*stride + i] with i < n_inner contiguous.
 * sign: -1 forward, +1 inverse (unnormalized).
 */
int nbk_fft_r2c_z(const double* real, double* cplx,
                  int64_t nlines, int64_t nz, double scale, void* stream);
int nbk_fft_c2r_z(const double* cplx, double* real,
                  int64_t nlines, int64_t nz, void* stream);
int nbk_fft_c_strided(double* cplx, int64_t nfft, int64_t stride,
                      int64_t n_outer, int64_t outer_stride,
                      int64_t n_inner, int sign, void* stream);

/* k-space elementwise ------------------------------------------------
 * All of these address a local complex slab of logical global shape
 * (n0, n1, n2h); `axis_map` gives, for each local axis, which global
 * axis it carries (identity (0,1,2) untransposed; (1,0,2) for the
 * post-alltoall transposed layout), and `off` the global start of the
 * local block along each LOCAL axis.  dims = local shape.
 */
int nbk_compensate_f64(double* cplx, const int64_t nmesh[3],
                       const int64_t dims[3], const int64_t off[3],
                       const int axis_map[3],
                       int window, int interlaced, void* stream);

int nbk_interlace_combine_f64(double* c1, const double* c2,
                              const int64_t nmesh[3], const double box[3],
                              const int64_t dims[3], const int64_t off[3],
                              const int axis_map[3], void* stream);

int nbk_power3d_f64(double* out, const double* c1, const double* c2,
                    double volume,
                    const int64_t dims[3], const int64_t off[3],
                    int clear_zero_mode, void* stream);

/* binning ------------------------------------------------------------
 * project_to_basis sums (fftpower.py:595-666) over the local slab:
 *   xsum, musum            : (Nx+2)*(Nmu+2) doubles
 *   Nsum                   : (Nx+2)*(Nmu+2) doubles (integral values)
 *   ysum                   : nell * (Nx+2)*(Nmu+2) complex doubles
 * kedges (device, nx_edges doubles) are digitized by binary search on
 * k^2; muedges (device, nmu_edges doubles) likewise on mu; Hermitian
 * double-count weights along global axis 2 (weight 2 where kz > 0,
 * meshtools.py:188-215).  `ells` lists the multipole orders (first
 * entry must be 0, mirroring fftpower.py:585).  los is the
 * line-of-sight unit vector.  Outputs are accumulated (caller zeroes).
 * kedges must hold SQUARED bin edges (the digitize runs on |k|^2 / r^2,
 * fftpower.py:578,615).  real_field = 1 bins a configuration-space
 * RealField instead (FFTCorr): relative-position coordinates, all axes
 * full length, no Hermitian double-count (fftcorr.py:150-176).
 */
int nbk_bin_power_f64(const double* cplx, const int64_t nmesh[3],
                      const double box[3],
                      const int64_t dims[3], const int64_t off[3],
                      const int axis_map[3],
                      const double* kedges, int64_t nx_edges,
                      const double* muedges, int64_t nmu_edges,
                      const double los[3],
                      const int* ells, int nell,
                      int real_field,
                      double* xsum, double* musum, double* Nsum,
                      double* ysum, void* stream);

/* fused compensate + cross-power + binning: one streaming pass
 * computing comp1(c1) * conj(comp2(c2)) * volume per element (zero mode
 * cleared but still binned, fftpower.py:114-128) and accumulating the
 * project_to_basis sums — replaces the nbk_compensate_f64 (x2) +
 * nbk_power3d_f64 + nbk_bin_power_f64 sequence without materializing
 * p3d.  window1/window2: NBK_WINDOW_* or -1 for no compensation; c2 may
 * equal c1 (auto power) or be NULL (alias of c1).  The compensation
 * factors are bit-identical to nbk_compensate_f64's. */
int nbk_power_bin_f64(const double* c1, const double* c2, double volume,
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
                      double* ysum, void* stream);

/* fused x-pass FFT + compensate + auto-power + binning: consumes the
 * PRE-x-pass complex field (z and y axes already transformed, pencil
 * transpose already applied when distributed: element (j, c) of the
 * x-line through flattened local (y, zh) column c lives at
 * data[j * n_inner + c]), runs the final strided x FFT per line in LDS
 * and feeds the results straight into the project_to_basis sums — the
 * finished complex field is never written to HBM and columns wholly
 * beyond the last k-edge are skipped before their loads.  Equivalent to
 * nbk_fft_c_strided(axis 0) + nbk_power_bin_f64(c1 == c2) with
 * clear_zero semantics; bin assignment is bit-identical, the x-FFT
 * element values are bit-identical to the unfused pass.  kedges are
 * SQUARED k edges (as in nbk_bin_power_f64); out_sums is the contiguous
 * [xsum|musum|Nsum|ysum] block of (nx_edges+1)*(nmu_edges+1) doubles
 * each (ysum: nell planar re/im pairs).  nmesh[0] must be a power of
 * two in [8, 4096]; fails with NBK_ERR_UNSUPPORTED when the histogram +
 * FFT tile exceed the 160 KiB LDS (callers fall back to the unfused
 * sequence). */
/* data2: NULL for a plain auto power; for an INTERLACED mesh the
 * half-cell-shifted paint's pre-x-pass field — both tiles are FFT'd
 * and combined as c = a/2 + b/2 exp(i k.H/2) before compensation and
 * binning.  In that mode the self-conjugate z planes (iz = 0 and, for
 * even n2, the Nyquist plane) are SKIPPED — their Hermitian projection
 * couples columns, so the caller handles those two planes with the
 * standalone kernels and the same out_sums buffer (all the binning
 * entry points accumulate). */
/* dig_hints = {k_lo, k_invd, mu_lo, mu_invd}: the uniform edge-grid
 * parameters (kedges = arange(k_lo, ., 1/k_invd), muedges =
 * linspace(mu_lo, ., 1/mu_invd)) used only as digitize STARTING
 * GUESSES — assignment is corrected against the exact edge arrays, so
 * it stays bit-identical to numpy.digitize. */
int nbk_fft_x_bin_f64(const double* data, const double* data2,
                      const int64_t nmesh[3],
                      int64_t n_inner, int64_t y_off,
                      const double box[3],
                      int window1, int interlaced1,
                      int clear_zero, double volume,
                      const double* kedges, int64_t nx_edges,
                      const double* muedges, int64_t nmu_edges,
                      const double dig_hints[4],
                      const double los[3],
                      const int* ells, int nell,
                      double* out_sums, void* stream);

/* small helpers ------------------------------------------------------ */
/* out[i] += a[i]  (f64, n elements) */
int nbk_axpy_f64(double* out, const double* a, double alpha, int64_t n,
                 void* stream);
/* mesh *= alpha  or mesh[:] = alpha when set_value != 0 */
int nbk_scale_f64(double* mesh, double alpha, int set_value, int64_t n,
                  void* stream);

#ifdef __cplusplus
}
#endif
#endif /* NBK_HIP_H */
