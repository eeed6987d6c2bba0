// k-space elementwise kernels: window compensation
// (nbodykit/source/mesh/catalog.py:419-594), interlaced-mesh combine
// (:341-347), and the 3D power p3d = c1 conj(c2) * V with the zero mode
// cleared (nbodykit/algorithms/fftpower.py:114-128).  All HBM-bound
// streaming passes (32 B/cell read-modify-write of c128); per-element
// trig is cheap VALU beside that.
#include "nbk_common.h"

namespace {

struct Layout {
    int64_t n0, n1, n2;        // global mesh
    int64_t d0, d1, d2;        // local dims
    int64_t o0, o1, o2;        // global offset of the local block
    int a0, a1, a2;            // which global axis each local axis carries
};

// global integer frequency triple (fx, fy, fz) of local flat index
__device__ __forceinline__ void global_freqs(const Layout& L, int64_t idx,
                                             double f[3]) {
    const int64_t l2 = idx % L.d2;
    const int64_t l1 = (idx / L.d2) % L.d1;
    const int64_t l0 = idx / (L.d2 * L.d1);
    int64_t g[3];
    g[L.a0] = l0 + L.o0;
    g[L.a1] = l1 + L.o1;
    g[L.a2] = l2 + L.o2;
    const int64_t n[3] = {L.n0, L.n1, L.n2};
    f[0] = freq_full(g[0], n[0]);
    f[1] = freq_full(g[1], n[1]);
    f[2] = freq_half(g[2], n[2]);   // compressed axis, Nyquist negative
}

__global__ void kcompensate(double* __restrict__ data, Layout L,
                            int window, int interlaced)
{
    const int64_t total = L.d0 * L.d1 * L.d2;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        double f[3];
        global_freqs(L, idx, f);
        const int64_t n[3] = {L.n0, L.n1, L.n2};
        // circular frequency w = 2 pi f / N in [-pi, pi)
        const double w[3] = {2.0 * M_PI * f[0] / (double)n[0],
                             2.0 * M_PI * f[1] / (double)n[1],
                             2.0 * M_PI * f[2] / (double)n[2]};
        const double corr = nbk_comp_factor(window, interlaced, w);
        data[2 * idx] *= corr;
        data[2 * idx + 1] *= corr;
    }
}

__global__ void kinterlace(double* __restrict__ c1,
                           const double* __restrict__ c2, Layout L)
{
    const int64_t total = L.d0 * L.d1 * L.d2;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        double f[3];
        global_freqs(L, idx, f);
        const int64_t n[3] = {L.n0, L.n1, L.n2};
        // k.H = sum_i 2 pi f_i / N_i; combine with phase exp(i k.H / 2)
        const double kH = 2.0 * M_PI * (f[0] / n[0] + f[1] / n[1]
                                        + f[2] / n[2]);
        double sp, cp;
        sincos(0.5 * kH, &sp, &cp);
        const cdouble a = {c1[2 * idx], c1[2 * idx + 1]};
        const cdouble b = {c2[2 * idx], c2[2 * idx + 1]};
        const cdouble ph = {cp, sp};
        const cdouble r = cadd(cscale(a, 0.5), cscale(cmul(b, ph), 0.5));
        c1[2 * idx] = r.re;
        c1[2 * idx + 1] = r.im;
    }
}

// FFTRecon displacement solve (fftrecon.py:222-238):
// out = i k_d / k^2 * exp(-k^2 R^2 / 2) / (bias (1 + (f/bias) mu^2)) * v
__global__ void krecon_disp(double* __restrict__ out,
                            const double* __restrict__ in, Layout L,
                            double k0x, double k0y, double k0z,
                            int axis, double R, double bias, double f,
                            double losx, double losy, double losz)
{
    const int64_t total = L.d0 * L.d1 * L.d2;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        double fr[3];
        global_freqs(L, idx, fr);
        const double kx = fr[0] * k0x, ky = fr[1] * k0y, kz = fr[2] * k0z;
        double k2 = kx * kx + ky * ky + kz * kz;
        const bool zero = (k2 == 0.0);
        if (zero) k2 = 1.0;
        const double mu = (kx * losx + ky * losy + kz * losz) / sqrt(k2);
        const double smooth = exp(-0.5 * k2 * R * R);
        const double frac = bias * (1.0 + f / bias * mu * mu);
        const double kd = (axis == 0) ? kx : (axis == 1) ? ky : kz;
        const double fac = zero ? 0.0 : kd / k2 * smooth / frac;
        const double re = in[2 * idx], im = in[2 * idx + 1];
        // multiply by i * fac
        out[2 * idx] = -im * fac;
        out[2 * idx + 1] = re * fac;
    }
}

__global__ void kpower3d(double* __restrict__ out,
                         const double* __restrict__ c1,
                         const double* __restrict__ c2,
                         double volume, int64_t total,
                         int64_t zero_idx /* local flat index or -1 */)
{
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        const cdouble a = {c1[2 * idx], c1[2 * idx + 1]};
        const cdouble b = {c2[2 * idx], c2[2 * idx + 1]};
        cdouble p = cmul(a, cconj(b));
        if (idx == zero_idx) p = {0.0, 0.0};
        out[2 * idx] = p.re * volume;
        out[2 * idx + 1] = p.im * volume;
    }
}

__global__ void kaxpy(double* __restrict__ out, const double* __restrict__ a,
                      double alpha, int64_t n)
{
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < n; i += stride)
        out[i] += alpha * a[i];
}

__global__ void kscale(double* __restrict__ m, double alpha, int set_value,
                       int64_t n)
{
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < n; i += stride)
        m[i] = set_value ? alpha : m[i] * alpha;
}

Layout make_layout(const int64_t nmesh[3], const int64_t dims[3],
                   const int64_t off[3], const int axis_map[3]) {
    Layout L;
    L.n0 = nmesh[0]; L.n1 = nmesh[1]; L.n2 = nmesh[2];
    L.d0 = dims[0]; L.d1 = dims[1]; L.d2 = dims[2];
    L.o0 = off[0]; L.o1 = off[1]; L.o2 = off[2];
    if (axis_map) { L.a0 = axis_map[0]; L.a1 = axis_map[1]; L.a2 = axis_map[2]; }
    else { L.a0 = 0; L.a1 = 1; L.a2 = 2; }
    return L;
}

int egrid(int64_t total, int block) {
    int64_t g = (total + block - 1) / block;
    if (g > 524288) g = 524288;
    if (g < 1) g = 1;
    return (int)g;
}

}  // namespace

extern "C" int nbk_compensate_f64(double* cplx, const int64_t nmesh[3],
                                  const int64_t dims[3], const int64_t off[3],
                                  const int axis_map[3],
                                  int window, int interlaced, void* stream)
{
    if (window < 0 || window > 2) {
        NBK_SET_ERR("nbk_compensate_f64: bad window %d", window);
        return NBK_ERR_ARG;
    }
    Layout L = make_layout(nmesh, dims, off, axis_map);
    const int64_t total = L.d0 * L.d1 * L.d2;
    hipLaunchKernelGGL(kcompensate, dim3(egrid(total, 256)), dim3(256), 0,
                       (hipStream_t)stream, cplx, L, window, interlaced);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_interlace_combine_f64(double* c1, const double* c2,
                                         const int64_t nmesh[3],
                                         const double box[3],
                                         const int64_t dims[3],
                                         const int64_t off[3],
                                         const int axis_map[3], void* stream)
{
    (void)box;   // k.H reduces to 2 pi f / N — box-independent
    Layout L = make_layout(nmesh, dims, off, axis_map);
    const int64_t total = L.d0 * L.d1 * L.d2;
    hipLaunchKernelGGL(kinterlace, dim3(egrid(total, 256)), dim3(256), 0,
                       (hipStream_t)stream, c1, c2, L);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_recon_displacement_f64(double* out, const double* in,
                                          const int64_t nmesh[3],
                                          const double box[3],
                                          const int64_t dims[3],
                                          const int64_t off[3],
                                          int axis, double R, double bias,
                                          double f, const double los[3],
                                          void* stream)
{
    if (axis < 0 || axis > 2) {
        NBK_SET_ERR("nbk_recon_displacement_f64: bad axis %d", axis);
        return NBK_ERR_ARG;
    }
    Layout L = make_layout(nmesh, dims, off, nullptr);
    const int64_t total = L.d0 * L.d1 * L.d2;
    hipLaunchKernelGGL(krecon_disp, dim3(egrid(total, 256)), dim3(256), 0,
                       (hipStream_t)stream, out, in, L,
                       2.0 * M_PI / box[0], 2.0 * M_PI / box[1],
                       2.0 * M_PI / box[2], axis, R, bias, f,
                       los[0], los[1], los[2]);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_power3d_f64(double* out, const double* c1,
                               const double* c2, double volume,
                               const int64_t dims[3], const int64_t off[3],
                               int clear_zero_mode, void* stream)
{
    const int64_t total = dims[0] * dims[1] * dims[2];
    // local flat index of the global (0,0,0) mode, if owned here
    int64_t zero_idx = -1;
    if (clear_zero_mode && off[0] == 0 && off[1] == 0 && off[2] == 0)
        zero_idx = 0;
    else if (clear_zero_mode) {
        // offsets are per-local-axis global starts; mode (0,0,0) is local
        // only when every offset is 0 (slab partitions always start a
        // block at the global origin on exactly one rank)
        zero_idx = -1;
    }
    hipLaunchKernelGGL(kpower3d, dim3(egrid(total, 256)), dim3(256), 0,
                       (hipStream_t)stream, out, c1, c2, volume, total,
                       zero_idx);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_axpy_f64(double* out, const double* a, double alpha,
                            int64_t n, void* stream)
{
    hipLaunchKernelGGL(kaxpy, dim3(egrid(n, 256)), dim3(256), 0,
                       (hipStream_t)stream, out, a, alpha, n);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}

extern "C" int nbk_scale_f64(double* mesh, double alpha, int set_value,
                             int64_t n, void* stream)
{
    hipLaunchKernelGGL(kscale, dim3(egrid(n, 256)), dim3(256), 0,
                       (hipStream_t)stream, mesh, alpha, set_value, n);
    NBK_CHECK_HIP(hipGetLastError());
    return NBK_OK;
}
