"""
FFTRecon restatement (reference nbodykit/algorithms/fftrecon.py:11-269):
Lagrangian reconstruction in a periodic box — paint delta_d, solve the
smoothed Zel'dovich displacement per axis in k-space
(i k_d/k^2 exp(-k^2 R^2/2) / (bias (1 + f/bias mu^2)), :222-238),
read it out at the particle positions (CIC, :246-249), shift data (and
optionally randoms) and combine with the LGS/LF2/LRR schemes
(:173-216).  The field convention is delta = counts/nbar (NOT 1+delta;
:163 and the FIXME at :215).
"""
import numpy

from .mesh import MeshGeometry, r2c, c2r
from .paint import paint, readout


def _work_with(pos, s, geom, nbar_count):
    mesh = numpy.zeros(tuple(int(x) for x in geom.Nmesh))
    dpos = pos if s is None else (pos - s)
    paint(dpos, 1.0, mesh, geom, resampler='cic')
    mesh /= nbar_count
    return mesh


def fftrecon_oracle(data_pos, ran_pos, Nmesh, BoxSize, bias=1.0, f=0.0,
                    los=(0, 0, 1), R=20.0, revert_rsd_random=False,
                    scheme='LGS'):
    """Returns the reconstructed real-space field (numpy array)."""
    assert scheme in ('LGS', 'LF2', 'LRR')
    geom = MeshGeometry(Nmesh, BoxSize, dtype='f8')
    los = numpy.asarray(los, dtype='f8')
    N = tuple(int(x) for x in geom.Nmesh)

    nbar_d = len(data_pos) / float(numpy.prod(geom.Nmesh))
    nbar_r = len(ran_pos) / float(numpy.prod(geom.Nmesh))

    # delta_d (unshifted) -> k-space (fftrecon.py:240-242)
    delta_d = _work_with(data_pos, None, geom, nbar_d)
    delta_k = r2c(delta_d, geom)

    # wavenumber grids (Nyquist negative)
    k0 = 2 * numpy.pi / geom.BoxSize
    fx = numpy.fft.fftfreq(N[0]) * N[0]
    fy = numpy.fft.fftfreq(N[1]) * N[1]
    fz = numpy.arange(N[2] // 2 + 1, dtype='f8')
    if N[2] % 2 == 0:
        fz[-1] = -(N[2] // 2)
    kx = (fx * k0[0]).reshape(-1, 1, 1)
    ky = (fy * k0[1]).reshape(1, -1, 1)
    kz = (fz * k0[2]).reshape(1, 1, -1)
    k2 = kx ** 2 + ky ** 2 + kz ** 2
    zero = k2 == 0
    k2s = numpy.where(zero, 1.0, k2)
    mu = (kx * los[0] + ky * los[1] + kz * los[2]) / numpy.sqrt(k2s)
    smooth = numpy.exp(-0.5 * k2s * R * R)
    frac = bias * (1 + f / bias * mu ** 2)

    def solve_displacement(pos):
        s = numpy.zeros((len(pos), 3))
        for d, kd in enumerate((kx, ky, kz)):
            disp_k = 1j * kd / k2s * smooth / frac * delta_k
            disp_k[zero] = 0
            disp = c2r(disp_k, geom)
            s[:, d] = readout(pos, disp, geom, resampler='cic')
        return s

    s_d = solve_displacement(data_pos)
    s_r = solve_displacement(ran_pos)

    # RSD conventions (fftrecon.py:260-267)
    s_d = s_d * (1 + los * f)
    if revert_rsd_random:
        s_r = s_r * (1 + los * f)

    delta_s_r = _work_with(ran_pos, s_r, geom, nbar_r)

    def LGS():
        delta_s_d = _work_with(data_pos, s_d, geom, nbar_d)
        return delta_s_d - delta_s_r

    def LRR():
        delta_s_nr = _work_with(ran_pos, -s_r, geom, nbar_r)
        delta_du = _work_with(data_pos, None, geom, nbar_d)
        return delta_du - 0.5 * (delta_s_nr + delta_s_r)

    if scheme == 'LGS':
        return LGS()
    if scheme == 'LRR':
        return LRR()
    return 3.0 / 7.0 * LGS() + 4.0 / 7.0 * LRR()
