"""Occupancy sweep for the strided FFT pass in BOTH step geometries
(y pass: stride nzh, and the legacy x geometry), over NBK_FFT_BLOCK x
NBK_FFT_TI.  More, smaller blocks = more independent load streams per
CU (the pass is latency-bound).

Usage (GPU box): python tests/debug_ypass.py
"""
import json
import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def one(reps=5):
    import torch
    from nbodykit_amd import hiplib
    lib = hiplib.require()
    n = 1024
    nzh = n // 2 + 1
    torch.manual_seed(3)
    data = (torch.randn(n, n, nzh, dtype=torch.float64, device='cuda')
            + 0j).to(torch.complex128).contiguous()

    def ypass():
        # axis-1 pass over view (n, n, nzh): nfft=n, stride=nzh,
        # n_outer=n, outer_stride=n*nzh, n_inner=nzh
        hiplib.check(lib.nbk_fft_c_strided(
            hiplib.dptr(data), n, nzh, n, n * nzh, nzh, -1,
            hiplib.cur_stream()), 'y')

    def xpass():
        hiplib.check(lib.nbk_fft_c_strided(
            hiplib.dptr(data), n, n * nzh, 1, n * n * nzh, n * nzh, -1,
            hiplib.cur_stream()), 'x')

    out = {'BLOCK': os.environ.get('NBK_FFT_BLOCK', '1024'),
           'TI': os.environ.get('NBK_FFT_TI', '4')}
    for name, fn in (('y', ypass), ('x', xpass)):
        fn()
        torch.cuda.synchronize()
        e0 = torch.cuda.Event(enable_timing=True)
        e1 = torch.cuda.Event(enable_timing=True)
        e0.record()
        for _ in range(reps):
            fn()
        e1.record()
        torch.cuda.synchronize()
        out[name + '_ms'] = round(e0.elapsed_time(e1) / reps, 3)
    print(json.dumps(out))


if __name__ == '__main__':
    if os.environ.get('NBK_CHILD'):
        one()
        sys.exit(0)
    env = dict(os.environ, NBK_CHILD='1')
    for blk, ti in (('1024', '4'), ('512', '4'), ('512', '2'),
                    ('256', '2'), ('256', '1'), ('1024', '2'),
                    ('1024', '8'), ('512', '8')):
        e = dict(env, NBK_FFT_BLOCK=blk, NBK_FFT_TI=ti)
        subprocess.run([sys.executable, os.path.abspath(__file__)],
                       env=e, check=False)
