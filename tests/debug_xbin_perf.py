"""Isolation timing for nbk_fft_x_bin_f64 vs the unfused
kfft_c_strided(x) + nbk_power_bin_f64 pair at C4 geometry (1024^3).

Usage (GPU box):  python tests/debug_xbin_perf.py [reps]
Env sweeps: NBK_XBIN_TI, NBK_XBIN_GRID (read once per process — the
sweep loop below re-execs itself per setting).
"""
import json
import os
import subprocess
import sys

import numpy
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def one_config(reps=3):
    from nbodykit_amd import hiplib
    lib = hiplib.require()
    n0 = n1 = n2 = int(os.environ.get('NBK_XBIN_N', '1024'))
    nzh = n2 // 2 + 1
    box = 5000.0
    torch.manual_seed(7)
    data = (torch.randn(n0, n1, nzh, dtype=torch.float64, device='cuda')
            + 1j * torch.randn(n0, n1, nzh, dtype=torch.float64,
                               device='cuda')) * 1e-3
    data = data.to(torch.complex128).contiguous()
    n_inner = n1 * nzh

    dk = 2 * numpy.pi / box
    kmax = numpy.pi * n0 / box + dk / 2
    kedges = numpy.arange(0.0, kmax, dk)
    muedges = numpy.linspace(-1, 1, 2, endpoint=True)
    k2e = torch.as_tensor(kedges ** 2).to('cuda')
    mue = torch.as_tensor(muedges).to('cuda')
    NB = (len(kedges) + 1) * (len(muedges) + 1)
    nfields = 3 + 2 * 1
    sums = torch.zeros(nfields * NB, dtype=torch.float64, device='cuda')
    nmesh = hiplib.i64_arr([n0, n1, n2])
    boxa = hiplib.f64_arr([box] * 3)
    los = hiplib.f64_arr([0., 0., 1.])
    ells = hiplib.int_arr([0])
    vol = box ** 3

    def run_xbin():
        hiplib.check(lib.nbk_fft_x_bin_f64(
            hiplib.dptr(data), None, nmesh, n_inner, 0, boxa,
            0, 0, 1, vol,
            hiplib.dptr(k2e), len(kedges), hiplib.dptr(mue), len(muedges),
            hiplib.f64_arr([0.0, 1.0 / dk, -1.0, 0.5]),
            los, ells, 1, hiplib.dptr(sums), hiplib.cur_stream()),
            'nbk_fft_x_bin_f64')

    work = data.clone()

    def run_xpass():
        hiplib.check(lib.nbk_fft_c_strided(
            hiplib.dptr(work), n0, n_inner, 1, n0 * n_inner, n_inner,
            -1, hiplib.cur_stream()), 'nbk_fft_c_strided')

    def run_kbin():
        hiplib.check(lib.nbk_power_bin_f64(
            hiplib.dptr(work), None, vol, 0, 0, 0, 0, 1,
            nmesh, boxa, hiplib.i64_arr([n0, n1, nzh]),
            hiplib.i64_arr([0, 0, 0]), None,
            hiplib.dptr(k2e), len(kedges), hiplib.dptr(mue), len(muedges),
            los, ells, 1,
            hiplib.dptr(sums), hiplib.dptr(sums[NB:]),
            hiplib.dptr(sums[2 * NB:]), hiplib.dptr(sums[3 * NB:]),
            hiplib.cur_stream()), 'nbk_power_bin_f64')

    def time_fn(fn, label):
        fn()  # warm
        torch.cuda.synchronize()
        ev0 = torch.cuda.Event(enable_timing=True)
        ev1 = torch.cuda.Event(enable_timing=True)
        ev0.record()
        for _ in range(reps):
            fn()
        ev1.record()
        torch.cuda.synchronize()
        ms = ev0.elapsed_time(ev1) / reps
        return ms

    t_xbin = time_fn(run_xbin, 'xbin')
    t_xpass = time_fn(run_xpass, 'xpass')
    t_kbin = time_fn(run_kbin, 'kbin')
    out = {'N': n0,
           'TI': os.environ.get('NBK_XBIN_TI', '4'),
           'GRID': os.environ.get('NBK_XBIN_GRID', '2048'),
           'PH': os.environ.get('NBK_XBIN_PHASES', '3'),
           'xbin_ms': round(t_xbin, 3),
           'xpass_ms': round(t_xpass, 3),
           'kbin_ms': round(t_kbin, 3),
           'unfused_ms': round(t_xpass + t_kbin, 3)}
    print(json.dumps(out))
    return out


if __name__ == '__main__':
    if os.environ.get('NBK_XBIN_CHILD'):
        one_config(int(sys.argv[1]) if len(sys.argv) > 1 else 3)
        sys.exit(0)
    env = dict(os.environ, NBK_XBIN_CHILD='1')
    # phase decomposition at the C4 geometry, then a 512^3 probe (the
    # smaller field spans ~8x fewer DRAM pages — a large per-byte speed
    # jump would implicate address-translation misses)
    for ph in ('0', '1', '2', '3'):
        e = dict(env, NBK_XBIN_TI='4', NBK_XBIN_GRID='4096',
                 NBK_XBIN_PHASES=ph)
        subprocess.run([sys.executable, os.path.abspath(__file__), '3'],
                       env=e, check=False)
    for n in ('512', '1024'):
        e = dict(env, NBK_XBIN_TI='4', NBK_XBIN_GRID='4096',
                 NBK_XBIN_N=n)
        subprocess.run([sys.executable, os.path.abspath(__file__), '3'],
                       env=e, check=False)
