"""
Gaussian / lognormal mock generation (reference nbodykit/mockmaker.py)
— the input generator for the bench configs (SURVEY §8a "inputs").

Whitenoise: pmesh's ``generate_whitenoise`` algorithm is not restatable
from the reference tree, so this module defines its OWN
rank-invariant seeded field: unit-variance real-space normal variates
drawn cell-by-cell through MPIRandomState, transformed with
``delta_k = r2c(noise) * sqrt(P(k) N^3 / V)`` — the exact equivalence the
reference documents at mockmaker.py:27-36 ("generating real-space normal
variates with unity variance, calling r2c() and dividing by N^3, since
the variance of the unnormalized complex FFT is N^3 sigma^2": our r2c
carries 1/N^3, so the unit-variance complex field is r2c(noise)*N^{3/2},
and scaling by sqrt(P/V) gives the target spectrum).  Oracle parity runs
feed byte-identical catalogs to both paths, so the generator being ours
does not loosen the parity bar.

This CPU generator computes the full mesh on every rank (deterministic,
rank-invariant) and emits only the particles of the rank's x-slab;
particle order is global cell order, matching the reference's
mpsort-by-cell-id invariance (mockmaker.py:338-345).  The GPU generator
used by bench.py at large N lives in bench.py itself.
"""
import numbers

import numpy

from nbodykit_amd.mpirng import MPIRandomState
from nbodykit_amd.comm import SerialComm


def _full_whitenoise(nmesh, seed):
    """Unit-variance real-space noise for the FULL mesh, identical on
    every rank (serial MPIRandomState chunk semantics)."""
    rng = MPIRandomState(SerialComm(), seed=seed,
                         size=int(numpy.prod(nmesh)))
    return rng.normal().reshape(tuple(int(n) for n in nmesh))


def gaussian_real_fields(nmesh, boxsize, linear_power, seed,
                         unitary_amplitude=False, inverted_phase=False,
                         compute_displacement=False):
    """
    delta(x) with spectrum P(k), and optionally the Zel'dovich
    displacement psi_i(k) = i k_i / k^2 delta(k)
    (reference mockmaker.py:7-210).  Full-mesh numpy, deterministic.
    """
    if not isinstance(seed, numbers.Integral):
        raise ValueError("the seed used to generate the linear field must "
                         "be an integer")
    nmesh = numpy.asarray(nmesh, dtype='i8')
    boxsize = numpy.asarray(boxsize, dtype='f8')
    N = tuple(int(n) for n in nmesh)
    V = float(numpy.prod(boxsize))
    Ntot = float(numpy.prod(nmesh))

    noise = _full_whitenoise(nmesh, seed)
    delta_k = numpy.fft.rfftn(noise) / Ntot          # our r2c convention
    # unit-variance complex field (see module docstring)
    delta_k *= Ntot ** 0.5

    if unitary_amplitude:
        mod = numpy.abs(delta_k)
        mod[mod == 0] = 1.0
        delta_k /= mod

    if inverted_phase:
        delta_k *= -1

    # wavenumber grids (Nyquist negative, meshtools.py:150-153)
    k0 = 2 * numpy.pi / boxsize
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
    k2[zero] = 1.0

    # scale to the target spectrum: x sqrt(P/V) (mockmaker.py:108-124)
    power = linear_power(numpy.sqrt(k2).ravel()).reshape(k2.shape)
    delta_k *= (power / V) ** 0.5
    delta_k[zero] = 0.0

    delta = numpy.fft.irfftn(delta_k, s=N, axes=(0, 1, 2)) * Ntot

    disp = None
    if compute_displacement:
        disp = []
        for ki in (kx, ky, kz):
            with numpy.errstate(invalid='ignore', divide='ignore'):
                disp_k = 1j * ki / k2 * delta_k
            disp_k[zero] = 0.0
            disp.append(numpy.fft.irfftn(disp_k, s=N, axes=(0, 1, 2))
                        * Ntot)
    return delta, disp


def lognormal_transform(density, bias=1.):
    """F(delta) = exp(b delta) normalized to unit mean
    (reference mockmaker.py:213-243)."""
    out = numpy.exp(bias * density)
    out /= out.mean(dtype='f8')
    return out


def poisson_sample_to_points(delta, displacement, nmesh, boxsize, nbar,
                             comm, bias=1., seed=None):
    """
    Poisson-sample the (full-mesh) delta/displacement fields to this
    rank's particles (reference mockmaker.py:246-359): lognormal
    transform with lagrangian bias b-1, per-cell Poisson counts through
    the collective MPIRandomState, particles at cell corners + uniform
    in-cell shift, displacement read out at the cell (nnb at corner
    positions).  Output order is global cell order (x-slab concatenated),
    the reference's mpsort-by-cell-id invariant.
    """
    nmesh = numpy.asarray(nmesh, dtype='i8')
    boxsize = numpy.asarray(boxsize, dtype='f8')
    H = boxsize / nmesh

    seed1, seed2 = numpy.random.RandomState(seed).randint(
        0, 0xfffffff, size=2)

    field = lognormal_transform(delta, bias=bias - 1.)
    overallmean = float(numpy.prod(H)) * nbar
    cellmean = field * overallmean

    # this rank's x-slab of cells (contiguous C-order block)
    ws, rank = comm.size, comm.rank
    n0 = int(nmesh[0])
    if n0 % ws != 0 and ws > 1:
        raise ValueError("Nmesh[0] must be divisible by the rank count")
    nx_l = n0 // ws
    x0 = nx_l * rank

    lam = cellmean[x0:x0 + nx_l].ravel()
    rng = MPIRandomState(comm, seed=int(seed1), size=lam.size)
    Npc = rng.poisson(lam=lam)
    Npc = numpy.int64(Npc + 0.5)

    # cell corners of the local slab, in C order
    ix, iy, iz = numpy.meshgrid(
        numpy.arange(x0, x0 + nx_l), numpy.arange(nmesh[1]),
        numpy.arange(nmesh[2]), indexing='ij')
    corners = numpy.empty((lam.size, 3), dtype='f8')
    corners[:, 0] = ix.ravel() * H[0]
    corners[:, 1] = iy.ravel() * H[1]
    corners[:, 2] = iz.ravel() * H[2]

    disp_cells = numpy.empty((lam.size, 3), dtype='f8')
    for i in range(3):
        disp_cells[:, i] = displacement[i][x0:x0 + nx_l].ravel()

    pos = corners.repeat(Npc, axis=0)
    disp = disp_cells.repeat(Npc, axis=0)

    rng_shift = MPIRandomState(comm, seed=int(seed2), size=len(pos))
    # reference quirk kept: the shift amplitude is H[2] on ALL axes
    # (mockmaker.py:350-351 reuses the loop variable i == ndim-1)
    in_cell_shift = rng_shift.uniform(0, H[2], itemshape=(3,))

    pos[...] += in_cell_shift
    pos[...] %= boxsize

    return pos, disp
