"""Bisect the 96^3 mode-count deviation: call the bin kernel on
successively smaller sub-blocks (dims/off) of a ones-valued complex
field and compare Nsum against the CPU digitize, then print the exact
lines and near-edge k^2 values involved.  Run on a GPU box."""
import os
import sys

import numpy

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from nbodykit_amd import hiplib

N, L = 96, 300.


def cpu_counts(d0, d1, d2, o0, o1, o2, x2):
    f = numpy.fft.fftfreq(N) * N
    fz = numpy.arange(N // 2 + 1, dtype='f8')
    fz[-1] = -(N // 2)
    k0 = 2 * numpy.pi / L
    cnt = numpy.zeros(len(x2) + 1, 'i8')
    for a in range(d0):
        kx = float(f[a + o0] * k0)
        for b in range(d1):
            ky = float(f[b + o1] * k0)
            sxy = kx * kx + ky * ky
            kz = fz[o2:o2 + d2] * k0
            k2 = sxy + kz * kz
            w = numpy.where(fz[o2:o2 + d2] > 0, 2, 1)
            keep = k2 < x2[-1]
            d = numpy.digitize(k2[keep], x2)
            numpy.add.at(cnt, d, w[keep])
    return cnt


def gpu_counts(dims, off, x2, muedges):
    lib = hiplib.require()
    d0, d1, d2 = dims
    field = torch.ones((d0, d1, d2), dtype=torch.complex128,
                       device='cuda')
    NB = (len(x2) + 1) * (len(muedges) + 1)
    sums = torch.zeros(3 * NB, dtype=torch.float64, device='cuda')
    k2t = torch.as_tensor(x2).to('cuda')
    met = torch.as_tensor(muedges).to('cuda')
    hiplib.check(lib.nbk_bin_power_f64(
        hiplib.dptr(field), hiplib.i64_arr([N, N, N]),
        hiplib.f64_arr([L, L, L]),
        hiplib.i64_arr(dims), hiplib.i64_arr(off), None,
        hiplib.dptr(k2t), len(x2), hiplib.dptr(met), len(muedges),
        hiplib.f64_arr([0., 0., 1.]), hiplib.int_arr([0]), 0, 0,
        hiplib.dptr(sums), hiplib.dptr(sums[NB:]),
        hiplib.dptr(sums[2 * NB:]), hiplib.dptr(sums[3 * NB:]),
        hiplib.cur_stream()), 'bin')
    torch.cuda.synchronize()
    Nsum = sums.cpu().numpy()[2 * NB:3 * NB]
    # Nsum layout: (Nx+2, Nmu+2); sum over mu columns -> per-kbin
    Nsum = Nsum.reshape(len(x2) + 1, len(muedges) + 1).sum(axis=1)
    # reconstruct digitize-indexed counts: kernel skips overflow
    cnt = numpy.zeros(len(x2) + 1, 'i8')
    cnt[:len(x2) + 1] = numpy.round(Nsum).astype('i8')
    return cnt


def main():
    k0 = 2 * numpy.pi / L
    dk = k0
    kedges = numpy.arange(0., numpy.pi * N / L + dk / 2, dk)
    x2 = kedges ** 2
    muedges = numpy.linspace(-1, 1, 2)

    g = gpu_counts((N, N, N // 2 + 1), (0, 0, 0), x2, muedges)
    c = cpu_counts(N, N, N // 2 + 1, 0, 0, 0, x2)
    bad = numpy.flatnonzero(g != c)
    print('full-field mismatched digitize bins:', bad, flush=True)
    print('gpu:', g[bad], 'cpu:', c[bad], flush=True)
    if not len(bad):
        print('NO MISMATCH (!!)')
        return

    bad_slabs = []
    for i in range(N):
        gs = gpu_counts((1, N, N // 2 + 1), (i, 0, 0), x2, muedges)
        cs = cpu_counts(1, N, N // 2 + 1, i, 0, 0, x2)
        if not numpy.array_equal(gs, cs):
            bad_slabs.append(i)
    print('mismatched slabs:', bad_slabs, flush=True)

    f = numpy.fft.fftfreq(N) * N
    shown = 0
    for i in bad_slabs[:4]:
        for j in range(N):
            gl = gpu_counts((1, 1, N // 2 + 1), (i, j, 0), x2, muedges)
            cl = cpu_counts(1, 1, N // 2 + 1, i, j, 0, x2)
            if not numpy.array_equal(gl, cl):
                db = numpy.flatnonzero(gl != cl)
                kx = float(f[i] * k0)
                ky = float(f[j] * k0)
                sxy = kx * kx + ky * ky
                fz = numpy.arange(N // 2 + 1, dtype='f8')
                fz[-1] = -(N // 2)
                kz = fz * k0
                k2 = sxy + kz * kz
                print('line (%d,%d) f=(%g,%g) bins %s gpu %s cpu %s'
                      % (i, j, f[i], f[j], db, gl[db], cl[db]),
                      flush=True)
                for b in db:
                    near = numpy.flatnonzero(
                        numpy.abs(k2 - x2[min(b, len(x2) - 1)])
                        < 1e-12 * max(1., x2[min(b, len(x2) - 1)]))
                    for g2 in near:
                        print('   fz=%g k2=%.20e edge[%d]=%.20e diff=%g'
                              % (fz[g2], k2[g2], b,
                                 x2[min(b, len(x2) - 1)],
                                 k2[g2] - x2[min(b, len(x2) - 1)]),
                              flush=True)
                shown += 1
                if shown > 6:
                    return


if __name__ == '__main__':
    main()
