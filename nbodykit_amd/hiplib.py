"""
ctypes binding of libnbk_hip.so (ABI: include/nbk_hip.h).

Loading the library works without a GPU (symbol checks run in CPU-only
CI); *compute* calls require a device.  The product compute path calls
:func:`require` first and raises RuntimeError if the extension or a GPU
is missing — there is no CPU fallback (DESIGN.md "Parity pinning").
"""
import ctypes
import os

import numpy

_LIBNAME = 'libnbk_hip.so'

WINDOW_IDS = {'cic': 0, 'tsc': 1, 'pcs': 2}

EXPORTED_SYMBOLS = [
    'nbk_version', 'nbk_last_error_string', 'nbk_device_count',
    'nbk_paint_f64', 'nbk_paint_sorted_f64', 'nbk_readout_f64', 'nbk_recon_displacement_f64',
    'nbk_bucket_count_f64', 'nbk_bucket_scatter_f64',
    'nbk_xsort_count_f64', 'nbk_xsort_scatter_f64',
    'nbk_bucket_fine_f64', 'nbk_scan_matrix_i32', 'nbk_paint_gather_f64',
    'nbk_psort_count_f64', 'nbk_psort_scatter_f64',
    'nbk_paint_gather_fft_f64',
    'nbk_fft_r2c_z', 'nbk_fft_c2r_z', 'nbk_fft_c_strided',
    'nbk_compensate_f64', 'nbk_interlace_combine_f64', 'nbk_power3d_f64',
    'nbk_bin_power_f64', 'nbk_power_bin_f64', 'nbk_fft_x_bin_f64',
    'nbk_axpy_f64', 'nbk_scale_f64',
]

_lib = None
_load_error = None

c_i64 = ctypes.c_int64
c_i64_p = ctypes.POINTER(ctypes.c_int64)
c_int_p = ctypes.POINTER(ctypes.c_int)
c_f64 = ctypes.c_double
c_f64_p = ctypes.POINTER(ctypes.c_double)
c_void = ctypes.c_void_p


def _declare(lib):
    lib.nbk_version.restype = ctypes.c_char_p
    lib.nbk_version.argtypes = []
    lib.nbk_last_error_string.restype = ctypes.c_char_p
    lib.nbk_last_error_string.argtypes = []
    lib.nbk_device_count.restype = ctypes.c_int
    lib.nbk_device_count.argtypes = []

    lib.nbk_paint_f64.restype = ctypes.c_int
    lib.nbk_paint_f64.argtypes = [c_void, c_void, c_i64, c_i64_p, c_f64_p,
                                  ctypes.c_int, c_f64, c_void, c_i64, c_i64,
                                  c_void]
    lib.nbk_bucket_count_f64.restype = ctypes.c_int
    lib.nbk_bucket_count_f64.argtypes = [c_void, c_i64, c_i64_p, c_f64_p,
                                         c_void, c_void, c_void]
    lib.nbk_bucket_scatter_f64.restype = ctypes.c_int
    lib.nbk_bucket_scatter_f64.argtypes = [c_void, c_void, c_i64, c_i64_p,
                                           c_f64_p, c_void, c_void, c_void,
                                           c_void]
    lib.nbk_xsort_count_f64.restype = ctypes.c_int
    lib.nbk_xsort_count_f64.argtypes = [c_void, c_i64, ctypes.c_int,
                                        c_i64_p, c_f64_p, ctypes.c_int,
                                        c_void, c_void, c_void]
    lib.nbk_xsort_scatter_f64.restype = ctypes.c_int
    lib.nbk_xsort_scatter_f64.argtypes = [c_void, c_void, c_i64,
                                          ctypes.c_int, c_i64_p, c_f64_p,
                                          ctypes.c_int, c_void, c_void,
                                          c_void, c_void]
    lib.nbk_scan_matrix_i32.restype = ctypes.c_int
    lib.nbk_scan_matrix_i32.argtypes = [c_void, c_i64, c_i64, c_void,
                                        c_void, c_void, c_void]
    lib.nbk_bucket_fine_f64.restype = ctypes.c_int
    lib.nbk_bucket_fine_f64.argtypes = [c_void, c_void, c_i64, c_i64_p,
                                        c_f64_p, ctypes.c_int, c_void,
                                        c_void, c_void, c_void,
                                        ctypes.c_int, c_void]
    lib.nbk_paint_gather_f64.restype = ctypes.c_int
    lib.nbk_paint_gather_f64.argtypes = [c_void, c_void, c_i64, c_i64_p,
                                         c_f64_p, ctypes.c_int, c_f64,
                                         c_void, c_void, c_i64, c_i64,
                                         ctypes.c_int, ctypes.c_int,
                                         c_void]
    lib.nbk_paint_gather_fft_f64.restype = ctypes.c_int
    lib.nbk_paint_gather_fft_f64.argtypes = [c_void, c_void, c_i64,
                                             c_i64_p, c_f64_p,
                                             ctypes.c_int, c_f64, c_void,
                                             c_void, c_i64, c_i64, c_f64,
                                             ctypes.c_int, c_void]
    lib.nbk_psort_count_f64.restype = ctypes.c_int
    lib.nbk_psort_count_f64.argtypes = [c_void, c_i64, ctypes.c_int,
                                        c_i64_p, c_f64_p, ctypes.c_int,
                                        ctypes.c_int, ctypes.c_int,
                                        c_void, c_void]
    lib.nbk_psort_scatter_f64.restype = ctypes.c_int
    lib.nbk_psort_scatter_f64.argtypes = [c_void, c_void, c_i64,
                                          ctypes.c_int, c_i64_p, c_f64_p,
                                          ctypes.c_int, ctypes.c_int,
                                          ctypes.c_int, c_void, c_i64,
                                          c_void, c_void, c_void]
    lib.nbk_paint_sorted_f64.restype = ctypes.c_int
    lib.nbk_paint_sorted_f64.argtypes = lib.nbk_paint_f64.argtypes
    lib.nbk_readout_f64.restype = ctypes.c_int
    lib.nbk_readout_f64.argtypes = [c_void, c_i64, c_i64_p, c_f64_p,
                                    ctypes.c_int, c_void, c_i64, c_i64,
                                    c_void, c_void]
    lib.nbk_recon_displacement_f64.restype = ctypes.c_int
    lib.nbk_recon_displacement_f64.argtypes = [c_void, c_void, c_i64_p,
                                               c_f64_p, c_i64_p, c_i64_p,
                                               ctypes.c_int, c_f64, c_f64,
                                               c_f64, c_f64_p, c_void]
    lib.nbk_fft_r2c_z.restype = ctypes.c_int
    lib.nbk_fft_r2c_z.argtypes = [c_void, c_void, c_i64, c_i64, c_f64, c_void]
    lib.nbk_fft_c2r_z.restype = ctypes.c_int
    lib.nbk_fft_c2r_z.argtypes = [c_void, c_void, c_i64, c_i64, c_void]
    lib.nbk_fft_c_strided.restype = ctypes.c_int
    lib.nbk_fft_c_strided.argtypes = [c_void, c_i64, c_i64, c_i64, c_i64,
                                      c_i64, ctypes.c_int, c_void]
    lib.nbk_compensate_f64.restype = ctypes.c_int
    lib.nbk_compensate_f64.argtypes = [c_void, c_i64_p, c_i64_p, c_i64_p,
                                       c_int_p, ctypes.c_int, ctypes.c_int,
                                       c_void]
    lib.nbk_interlace_combine_f64.restype = ctypes.c_int
    lib.nbk_interlace_combine_f64.argtypes = [c_void, c_void, c_i64_p,
                                              c_f64_p, c_i64_p, c_i64_p,
                                              c_int_p, c_void]
    lib.nbk_power3d_f64.restype = ctypes.c_int
    lib.nbk_power3d_f64.argtypes = [c_void, c_void, c_void, c_f64, c_i64_p,
                                    c_i64_p, ctypes.c_int, c_void]
    lib.nbk_bin_power_f64.restype = ctypes.c_int
    lib.nbk_bin_power_f64.argtypes = [c_void, c_i64_p, c_f64_p, c_i64_p,
                                      c_i64_p, c_int_p, c_void, c_i64,
                                      c_void, c_i64, c_f64_p, c_int_p,
                                      ctypes.c_int, ctypes.c_int, c_void,
                                      c_void, c_void, c_void, c_void]
    lib.nbk_power_bin_f64.restype = ctypes.c_int
    lib.nbk_power_bin_f64.argtypes = [c_void, c_void, c_f64,
                                      ctypes.c_int, ctypes.c_int,
                                      ctypes.c_int, ctypes.c_int,
                                      ctypes.c_int,
                                      c_i64_p, c_f64_p, c_i64_p, c_i64_p,
                                      c_int_p, c_void, c_i64,
                                      c_void, c_i64, c_f64_p, c_int_p,
                                      ctypes.c_int,
                                      c_void, c_void, c_void, c_void,
                                      c_void]
    lib.nbk_fft_x_bin_f64.restype = ctypes.c_int
    lib.nbk_fft_x_bin_f64.argtypes = [c_void, c_void, c_i64_p, c_i64,
                                      c_i64, c_f64_p,
                                      ctypes.c_int, ctypes.c_int,
                                      ctypes.c_int, c_f64,
                                      c_void, c_i64, c_void, c_i64,
                                      c_f64_p,
                                      c_f64_p, c_int_p, ctypes.c_int,
                                      c_void, c_void]
    lib.nbk_axpy_f64.restype = ctypes.c_int
    lib.nbk_axpy_f64.argtypes = [c_void, c_void, c_f64, c_i64, c_void]
    lib.nbk_scale_f64.restype = ctypes.c_int
    lib.nbk_scale_f64.argtypes = [c_void, c_f64, ctypes.c_int, c_i64, c_void]


def load():
    """Load (or return the cached) library handle; raises on failure."""
    global _lib, _load_error
    if _lib is not None:
        return _lib
    if _load_error is not None:
        raise _load_error
    path = os.path.join(os.path.dirname(os.path.abspath(__file__)), _LIBNAME)
    try:
        if not os.path.exists(path):
            raise OSError("%s not found — build it with "
                          "`make -C nbodykit_amd/csrc` (or run "
                          "__graft_entry__.build())" % path)
        lib = ctypes.CDLL(path)
        _declare(lib)
    except OSError as exc:
        _load_error = RuntimeError(
            "nbodykit_amd HIP extension unavailable: %s" % exc)
        raise _load_error
    _lib = lib
    return lib


def available():
    try:
        load()
        return True
    except RuntimeError:
        return False


def require():
    """The compute-path gate: extension loaded AND a GPU present."""
    lib = load()
    import torch
    if not torch.cuda.is_available() or lib.nbk_device_count() < 1:
        raise RuntimeError(
            "nbodykit_amd requires an AMD GPU for the compute path "
            "(torch.cuda unavailable or no HIP device). There is no CPU "
            "fallback — use the oracle only for testing.")
    return lib


def check(rc, name):
    if rc != 0:
        lib = load()
        raise RuntimeError("%s failed (%d): %s"
                           % (name, rc, lib.nbk_last_error_string().decode()))


# ---- ctypes argument helpers -------------------------------------------

def i64_arr(values):
    return (ctypes.c_int64 * len(values))(*[int(v) for v in values])


def f64_arr(values):
    return (ctypes.c_double * len(values))(*[float(v) for v in values])


def int_arr(values):
    return (ctypes.c_int * len(values))(*[int(v) for v in values])


def dptr(tensor):
    """Device pointer of a torch tensor (or 0 for None)."""
    if tensor is None:
        return None
    return ctypes.c_void_p(tensor.data_ptr())


def cur_stream():
    import torch
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)
