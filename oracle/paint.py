"""
Window-deposit ("paint") restatement.

The reference calls pmesh's Cython scatter (``pm.paint`` at
nbodykit/source/mesh/catalog.py:287) with the standard particle-mesh
windows; the window shapes are fixed in-tree by the Fourier-space
compensation duals (source/mesh/catalog.py:453-594 = Jing 2005 eq. 18
with p = 2/3/4), i.e. the real-space kernels are the B-splines:

- cic (support 2): W(s) = 1-|s|, |s| < 1           [deposits to 2^3 cells]
- tsc (support 3): W(s) = 3/4-s^2 (|s|<1/2), (3/2-|s|)^2/2 (|s|<3/2)
- pcs (support 4): W(s) = (4-6s^2+3|s|^3)/6 (|s|<1), (2-|s|)^3/6 (|s|<2)

Alignment: a particle exactly on a grid point deposits its full mass
there (standard Hockney & Eastwood convention).  The interlaced second
mesh is painted at u + 0.5 in mesh units (``pm.affine.shift(0.5)``,
source/mesh/catalog.py:292); the k-space combine phase
exp(+0.5 i k.H) at :347 fixes this sign.
"""
import numpy

WINDOW_SUPPORT = {'cic': 2, 'tsc': 3, 'pcs': 4}


def _cic_offsets_weights(u):
    i0 = numpy.floor(u)
    frac = u - i0
    for dx in (0, 1):
        wx = frac[:, 0] if dx else 1.0 - frac[:, 0]
        for dy in (0, 1):
            wy = frac[:, 1] if dy else 1.0 - frac[:, 1]
            for dz in (0, 1):
                wz = frac[:, 2] if dz else 1.0 - frac[:, 2]
                yield (i0[:, 0] + dx, i0[:, 1] + dy, i0[:, 2] + dz,
                       wx * wy * wz)


def _tsc_w(s):
    a = numpy.abs(s)
    return numpy.where(a < 0.5, 0.75 - s * s,
                       numpy.where(a < 1.5, 0.5 * (1.5 - a) ** 2, 0.0))


def _pcs_w(s):
    a = numpy.abs(s)
    return numpy.where(a < 1.0, (4.0 - 6.0 * a * a + 3.0 * a ** 3) / 6.0,
                       numpy.where(a < 2.0, (2.0 - a) ** 3 / 6.0, 0.0))


def _centered_offsets_weights(u, offsets, wfunc, nearest):
    if nearest:
        base = numpy.floor(u + 0.5)       # tsc: centered on nearest point
    else:
        base = numpy.floor(u)             # pcs: centered between points
    for dx in offsets:
        wx = wfunc(u[:, 0] - (base[:, 0] + dx))
        for dy in offsets:
            wy = wfunc(u[:, 1] - (base[:, 1] + dy))
            for dz in offsets:
                wz = wfunc(u[:, 2] - (base[:, 2] + dz))
                yield (base[:, 0] + dx, base[:, 1] + dy, base[:, 2] + dz,
                       wx * wy * wz)


def paint(position, mass, mesh, geom, resampler='cic', shift=0.0):
    """
    Accumulate ``mass`` (scalar or per-particle) into ``mesh`` (modified
    in place, shape geom.Nmesh, f8) with periodic wrapping.  ``shift`` is
    in mesh units (0.5 for the interlaced second mesh).
    """
    if len(position) == 0:
        return mesh
    u = numpy.asarray(position, dtype='f8') / geom.H + shift
    m = numpy.broadcast_to(numpy.asarray(mass, dtype='f8'), (len(u),))

    if resampler == 'cic':
        gen = _cic_offsets_weights(u)
    elif resampler == 'tsc':
        gen = _centered_offsets_weights(u, (-1, 0, 1), _tsc_w, nearest=True)
    elif resampler == 'pcs':
        gen = _centered_offsets_weights(u, (-1, 0, 1, 2), _pcs_w, nearest=False)
    else:
        raise ValueError("unknown resampler '%s' (cic/tsc/pcs)" % resampler)

    N = geom.Nmesh
    for ix, iy, iz, w in gen:
        ix = numpy.remainder(ix.astype('i8'), N[0])
        iy = numpy.remainder(iy.astype('i8'), N[1])
        iz = numpy.remainder(iz.astype('i8'), N[2])
        numpy.add.at(mesh, (ix, iy, iz), w * m)
    return mesh


def readout(position, mesh, geom, resampler='cic'):
    """Windowed gather — the dual of :func:`paint` (pmesh ``readout``;
    FFTRecon reads displacements with the default CIC window,
    fftrecon.py:246-249; the LogNormal generator uses 'nnb',
    mockmaker.py:317-319)."""
    position = numpy.asarray(position, dtype='f8')
    if len(position) == 0:
        return numpy.zeros(0)
    u = position / geom.H
    N = geom.Nmesh
    out = numpy.zeros(len(u))

    if resampler == 'nnb':
        ix = numpy.remainder(numpy.floor(u[:, 0] + 0.5).astype('i8'), N[0])
        iy = numpy.remainder(numpy.floor(u[:, 1] + 0.5).astype('i8'), N[1])
        iz = numpy.remainder(numpy.floor(u[:, 2] + 0.5).astype('i8'), N[2])
        return mesh[ix, iy, iz].astype('f8')

    if resampler == 'cic':
        gen = _cic_offsets_weights(u)
    elif resampler == 'tsc':
        gen = _centered_offsets_weights(u, (-1, 0, 1), _tsc_w, nearest=True)
    elif resampler == 'pcs':
        gen = _centered_offsets_weights(u, (-1, 0, 1, 2), _pcs_w,
                                        nearest=False)
    else:
        raise ValueError("unknown resampler '%s'" % resampler)

    for ix, iy, iz, w in gen:
        ix = numpy.remainder(ix.astype('i8'), N[0])
        iy = numpy.remainder(iy.astype('i8'), N[1])
        iz = numpy.remainder(iz.astype('i8'), N[2])
        out += w * mesh[ix, iy, iz]
    return out
