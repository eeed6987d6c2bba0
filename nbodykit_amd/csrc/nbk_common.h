// Shared helpers for the nbk HIP kernels (gfx950 / CDNA4 only).
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <cmath>

#include "../../include/nbk_hip.h"

// ---- error plumbing ---------------------------------------------------
extern char nbk_errbuf[1024];

#define NBK_SET_ERR(...) snprintf(nbk_errbuf, sizeof(nbk_errbuf), __VA_ARGS__)

#define NBK_CHECK_HIP(expr)                                               \
    do {                                                                  \
        hipError_t _e = (expr);                                           \
        if (_e != hipSuccess) {                                           \
            NBK_SET_ERR("%s failed: %s (%s:%d)", #expr,                   \
                        hipGetErrorString(_e), __FILE__, __LINE__);       \
            return NBK_ERR_HIP;                                           \
        }                                                                 \
    } while (0)

// ---- device math helpers ---------------------------------------------
struct cdouble { double re, im; };

__device__ __forceinline__ cdouble cmul(cdouble a, cdouble b) {
    return {a.re * b.re - a.im * b.im, a.re * b.im + a.im * b.re};
}
__device__ __forceinline__ cdouble cadd(cdouble a, cdouble b) {
    return {a.re + b.re, a.im + b.im};
}
__device__ __forceinline__ cdouble csub(cdouble a, cdouble b) {
    return {a.re - b.re, a.im - b.im};
}
__device__ __forceinline__ cdouble cconj(cdouble a) { return {a.re, -a.im}; }
__device__ __forceinline__ cdouble cscale(cdouble a, double s) {
    return {a.re * s, a.im * s};
}

// periodic wrap of a (possibly negative) cell index.  Meshes are powers
// of two in this build (the FFT requires it), so the wrap is almost
// always a single AND — the general int64 modulo is a ~30-op sequence
// and the paint/count kernels issue up to 14 wraps per particle.  The
// n is wave-uniform, so the check is scalar and predicted.
__device__ __forceinline__ int64_t wrap_idx(int64_t i, int64_t n) {
    if (__builtin_expect((n & (n - 1)) == 0, 1))
        return i & (n - 1);
    i %= n;
    return i < 0 ? i + n : i;
}

// signed frequency of global index g on an axis of size n (numpy
// fftfreq order: Nyquist negative; meshtools.py:150-153).  The value
// reproduces numpy's ``fftfreq(n) * n`` BIT-EXACTLY, which computes
// (m * fl(1/n)) * n — a reciprocal multiply, NOT a division.  At
// non-power-of-two n this is NOT the exact integer (fftfreq(96)[7]*96
// = 7 -+ 1 ulp); the Python/oracle coordinate arrays carry that
// rounding, and modes sitting exactly on a k-bin edge digitize by it
// (verified: the reciprocal form matches fftfreq*n for every m at
// n = 24/48/96/100/640/1536/2048; plain division does not).  For
// power-of-two n both forms are the exact integer.
__device__ __forceinline__ double freq_full(int64_t g, int64_t n) {
    // positive frequencies run to (n-1)/2 inclusive — for even n that
    // is n/2 - 1 with the Nyquist stored negative, for odd n there is
    // no Nyquist plane (numpy fftfreq at any parity)
    const int64_t m = (g < n - n / 2) ? g : g - n;
    return ((double)m * (1.0 / (double)n)) * (double)n;
}
// compressed (last) axis: indices 0..n/2, Nyquist stored negative for
// EVEN n (odd n has no self-conjugate Nyquist plane and every g > 0 is
// a positive frequency).  These come from ``arange`` (+ an integer
// Nyquist assignment) on the Python side, so they ARE exact integers
// at any n.
__device__ __forceinline__ double freq_half(int64_t g, int64_t n) {
    return (double)((n % 2 == 0 && g == n / 2) ? -(n / 2) : g);
}

// Block barrier that waits only on LDS traffic (lgkmcnt), NOT on
// in-flight global loads (vmcnt) — plain __syncthreads emits
// s_waitcnt vmcnt(0) before s_barrier, which would force any
// software-pipelined global loads to complete at the first barrier
// they cross.  Safe whenever the only cross-thread state the barrier
// orders is LDS (all the FFT kernels: global reads are consumed
// through registers, whose use carries its own vmcnt wait).
__device__ __forceinline__ void nbk_sync_lds() {
    asm volatile("s_waitcnt lgkmcnt(0)\n\ts_barrier" ::: "memory");
}

// bit-reversal for the in-tile FFT (shared with nbk_fft.hip's local copy)
__device__ __forceinline__ int nbk_bitrev(int j, int bits) {
    return (int)(__brev((unsigned)j) >> (32 - bits));
}

// cross-TU access to nbk_fft.hip's cached device twiddle tables
// (W_N^j for j = 0..N/2, defined in nbk_fft.hip)
double* nbk_internal_twiddles(int64_t N);

// Inverse-window compensation factor for one element (the six filters of
// nbodykit/source/mesh/catalog.py:453-594).  w[i] = 2 pi f_i / N_i is the
// circular frequency in [-pi, pi); interlaced selects the plain
// 1/sinc^p inverse (Jing 2005 eq. 18), otherwise the first-order
// aliasing-corrected eq. 20 forms.  Shared by the standalone
// nbk_compensate_f64 pass and the fused nbk_power_bin_f64 so both paths
// are bit-identical.
// single-axis factor of the separable compensation: nbk_comp_factor is
// the product of three of these (as sequential divides; the fast bin
// path composes them as multiplies — a last-ulp difference covered by
// the 1e-10 fused-vs-unfused tolerance)
__device__ __forceinline__ double nbk_comp_factor1(int window,
                                                   int interlaced,
                                                   double wi) {
    if (interlaced) {
        const double s = 0.5 * wi;
        const double sc = (s == 0.0) ? 1.0 : sin(s) / s;
        const double p = (window == NBK_WINDOW_CIC) ? sc * sc
                       : (window == NBK_WINDOW_TSC) ? sc * sc * sc
                                                    : sc * sc * sc * sc;
        return 1.0 / p;
    }
    const double s2 = sin(0.5 * wi) * sin(0.5 * wi);
    double d;
    if (window == NBK_WINDOW_CIC)
        d = 1.0 - 2.0 / 3.0 * s2;
    else if (window == NBK_WINDOW_TSC)
        d = 1.0 - s2 + 2.0 / 15.0 * s2 * s2;
    else
        d = 1.0 - 4.0 / 3.0 * s2 + 2.0 / 5.0 * s2 * s2
            - 4.0 / 315.0 * s2 * s2 * s2;
    return 1.0 / sqrt(d);
}

__device__ __forceinline__ double nbk_comp_factor(int window, int interlaced,
                                                  const double w[3]) {
    double corr = 1.0;
    #pragma unroll
    for (int i = 0; i < 3; i++) {
        if (interlaced) {
            const double s = 0.5 * w[i];
            const double sc = (s == 0.0) ? 1.0 : sin(s) / s;
            const double p = (window == NBK_WINDOW_CIC) ? sc * sc
                           : (window == NBK_WINDOW_TSC) ? sc * sc * sc
                                                        : sc * sc * sc * sc;
            corr /= p;
        } else {
            const double s2 = sin(0.5 * w[i]) * sin(0.5 * w[i]);
            double d;
            if (window == NBK_WINDOW_CIC)
                d = 1.0 - 2.0 / 3.0 * s2;
            else if (window == NBK_WINDOW_TSC)
                d = 1.0 - s2 + 2.0 / 15.0 * s2 * s2;
            else
                d = 1.0 - 4.0 / 3.0 * s2 + 2.0 / 5.0 * s2 * s2
                    - 4.0 / 315.0 * s2 * s2 * s2;
            corr /= sqrt(d);
        }
    }
    return corr;
}
