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

// periodic wrap of a (possibly negative) cell index
__device__ __forceinline__ int64_t wrap_idx(int64_t i, int64_t n) {
    i %= n;
    return i < 0 ? i + n : i;
}

// signed integer frequency of global index g on an axis of size n
// (numpy fftfreq order: Nyquist negative; meshtools.py:150-153)
__device__ __forceinline__ double freq_full(int64_t g, int64_t n) {
    return (double)(g < n / 2 ? g : g - n);
}
// compressed (last) axis: indices 0..n/2, Nyquist stored negative
__device__ __forceinline__ double freq_half(int64_t g, int64_t n) {
    return (double)(g == n / 2 ? -(n / 2) : g);
}
