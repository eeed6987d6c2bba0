"""
Mesh geometry, coordinates, and the normalized FFT pair.

Conventions restated from in-tree evidence (pmesh itself is third-party and
absent):

- forward r2c is normalized by 1/Nmesh^3 so the complex field is
  dimensionless and the k=0 mode equals the configuration-space mean
  (nbodykit/algorithms/fftpower.py:126-128, nbodykit/mockmaker.py:27-36);
- c2r is the unnormalized inverse so c2r(r2c(x)) == x;
- the compressed (last) axis stores the half-spectrum, and the Nyquist
  frequency is represented as NEGATIVE in the coordinate arrays
  (nbodykit/meshtools.py:150-153; pinned by the ported Hermitian-weight
  test, nbodykit/tests/test_meshtools.py:67-93).
"""
import numpy


class MeshGeometry(object):
    """Nmesh/BoxSize bookkeeping (the non-MPI core of pmesh.pm.ParticleMesh
    as used at nbodykit/base/mesh.py:50)."""

    def __init__(self, Nmesh, BoxSize, dtype='f8'):
        _N = numpy.empty(3, dtype='i8')
        _N[:] = Nmesh
        _L = numpy.empty(3, dtype='f8')
        _L[:] = BoxSize
        self.Nmesh = _N
        self.BoxSize = _L
        self.dtype = numpy.dtype(dtype)

    @property
    def H(self):
        """cell size per dimension"""
        return self.BoxSize / self.Nmesh

    @property
    def cshape(self):
        """shape of the compressed complex field"""
        return (int(self.Nmesh[0]), int(self.Nmesh[1]),
                int(self.Nmesh[2]) // 2 + 1)


def _int_freqs(N):
    """Integer frequencies 0..N/2-1, -N/2..-1 (numpy fftfreq order;
    Nyquist negative, matching meshtools.py:150-153)."""
    return numpy.fft.fftfreq(N) * N


def complex_coords(geom):
    """Wavenumber coordinate list for the compressed complex field,
    shaped [(Nx,1,1), (1,Ny,1), (1,1,Nz/2+1)] for broadcasting
    (the ``.x`` attribute consumed by fftpower.py:570-605)."""
    Nx, Ny, Nz = (int(n) for n in geom.Nmesh)
    fx = _int_freqs(Nx)
    fy = _int_freqs(Ny)
    fz = numpy.arange(Nz // 2 + 1, dtype='f8')
    if Nz % 2 == 0:
        fz[-1] = -(Nz // 2)                  # Nyquist as negative (even)
    k0 = 2 * numpy.pi / geom.BoxSize
    return [
        (fx * k0[0]).reshape(Nx, 1, 1),
        (fy * k0[1]).reshape(1, Ny, 1),
        (fz * k0[2]).reshape(1, 1, Nz // 2 + 1),
    ]


def complex_circular_coords(geom):
    """'circular' frequency coordinates w = k*H in [-pi, pi)
    (nbodykit/base/mesh.py:136-140: the kind used by compensation
    filters, source/mesh/catalog.py:451)."""
    x = complex_coords(geom)
    H = geom.H
    return [xi * H[i] for i, xi in enumerate(x)]


def real_coords(geom):
    """Configuration 'relative' coordinates in [-L/2, L/2)
    (nbodykit/base/mesh.py:144)."""
    out = []
    for i in range(3):
        N = int(geom.Nmesh[i])
        c = _int_freqs(N) * geom.BoxSize[i] / N
        shape = [1, 1, 1]
        shape[i] = N
        out.append(c.reshape(shape))
    return out


def r2c(real, geom=None):
    """Forward real-to-complex, normalized by 1/N^3
    (evidence cited in module docstring)."""
    return numpy.fft.rfftn(numpy.asarray(real, dtype='f8')) / real.size


def c2r(cplx, geom):
    """Unnormalized inverse (so r2c . c2r == identity)."""
    shape = tuple(int(n) for n in geom.Nmesh)
    return numpy.fft.irfftn(cplx, s=shape, axes=(0, 1, 2)) \
        * float(numpy.prod(shape))
