"""
ZeldovichPower — the matter power spectrum in the Zel'dovich
approximation (reference nbodykit/cosmology/power/zeldovich.py:27-240).
The reference's mcfit integrals are restated on the biased FFTLog core
in correlation.py (mcfit is absent here; the normalizations below were
pinned against direct 2D quadrature of the Zel'dovich integral — see
tests/test_correlation_cpu.py — and C = 4 pi reproduced it to 1e-4):

    sigma_v^2 = 1/(6 pi^2) int dk P_L
    I0(r) = 1/(2 pi^2)     int dk P_L(k) j_0(kr)
    I1(r) = 1/(2 pi^2 r)   int dk/k P_L(k) j_1(kr)
    X(r)  = -2 I1 + 2 sigma_v^2,   Y(r) = -2 I0 + 6 I1
    I(k,n)= 4 pi int dr r^{2-n} f_n(r) j_n(kr)
    P_zel(k) = sum_n I(k, n) with
      f_0 = exp(-k^2 (X+Y)/2) - exp(-k^2 sigma_v^2)
      f_n = (k Y)^n exp(-k^2 (X+Y)/2)            (n >= 1)
and the low-k expansion
    P(k) = (1 - k^2 sv^2 + k^4 sv^4 / 2) P_L + Q3/2,
    Q3 = k^4/(10 pi^2) int dq P_L^2/q^2
below k = 5e-3 (reference :123-151).
"""
import numpy
from scipy.integrate import quad
from scipy.interpolate import InterpolatedUnivariateSpline as spline

from .correlation import _mellin_sph_bessel

NUM_PTS = 1024
KMIN = 1e-5
KMAX = 1e2


class ZeldovichPower(object):

    def __init__(self, cosmo, redshift, transfer='EisensteinHu',
                 nmax=32):
        from nbodykit_amd.cosmology import LinearPower
        self.Plin = LinearPower(cosmo, redshift, transfer=transfer)
        self.nmax = nmax
        self.cosmo = cosmo
        self._sigma8 = getattr(cosmo, 'sigma8', None)
        self._z = redshift

        self._k0_low = 5e-3
        self._attrs = {}
        self._attrs.update(getattr(self.Plin, 'attrs', {}))
        self._setup()

    @property
    def attrs(self):
        self._attrs['redshift'] = self.redshift
        self._attrs['sigma8'] = self.sigma8
        return self._attrs

    def _setup(self):
        k = numpy.logspace(numpy.log10(KMIN), numpy.log10(KMAX),
                           NUM_PTS)
        Pk = self.Plin(k)

        # I0/I1 on the reflected r grid (quadrature-pinned constants)
        r, G0 = _mellin_sph_bessel(k, k * Pk, 0, q=1.0)
        _, G1 = _mellin_sph_bessel(k, Pk, 1, q=0.0)
        I0 = G0 / (2 * numpy.pi ** 2)
        I1 = G1 / (2 * numpy.pi ** 2) / r

        self._sigmasq = self.Plin.velocity_dispersion(
            kmin=1e-5, kmax=10., limit=500) ** 2
        X = -2. * I1 + 2 * self._sigmasq
        Y = -2. * I0 + 6. * I1
        # the FFTLog edges of I0/I1 ring (no low-ringing phase); trim
        # them — r^2 weighting makes the dropped small-r band
        # negligible in the power integral (validated vs quadrature)
        sl = slice(NUM_PTS // 8, -NUM_PTS // 8)
        self._r = r[sl]
        self._X = X[sl]
        self._Y = Y[sl]

        self._Q3 = quad(lambda q: (self.Plin(q) / q) ** 2, 1e-6,
                        100.)[0]

    @property
    def redshift(self):
        return self._z

    @redshift.setter
    def redshift(self, value):
        self._z = value
        self.Plin.redshift = value
        self._setup()

    @property
    def sigma8(self):
        return self._sigma8

    @sigma8.setter
    def sigma8(self, value):
        self._sigma8 = value
        self.Plin.sigma8 = value
        self._setup()

    def _low_k_approx(self, k):
        Q3 = 1. / (10. * numpy.pi ** 2) * k ** 4 * self._Q3
        Plin = self.Plin(k)
        term1 = (1 - k ** 2 * self._sigmasq
                 + 0.5 * k ** 4 * self._sigmasq ** 2) * Plin
        return term1 + 0.5 * Q3

    def __call__(self, k):
        """P_zel(k) in (Mpc/h)^3; k in h/Mpc."""
        k = numpy.asarray(k, dtype='f8')
        scalar = k.ndim == 0
        k = numpy.atleast_1d(k)
        out = numpy.zeros_like(k)

        low = k < self._k0_low
        if low.any():
            out[low] = self._low_k_approx(k[low])

        hi = ~low
        if hi.any():
            khi = k[hi]
            Pzel = numpy.zeros_like(khi)
            r = self._r
            # spline each order's integral once; every k shares it
            # (reference rebuilds the mcfit object per k — same math)
            for n in range(0, self.nmax + 1):
                kmax_fac = numpy.exp(-0.5 * numpy.outer(khi ** 2,
                                                        self._X + self._Y))
                if n > 0:
                    f = (numpy.outer(khi, self._Y)) ** n * kmax_fac
                else:
                    f = kmax_fac - numpy.exp(
                        -numpy.outer(khi ** 2, [self._sigmasq]))
                # I(k, n) = (2 pi)^{3/2} int dr/r r^{3-n} f(r) j_n(kr):
                # f depends on k, so transform per k row — but the
                # FFTLog gives I at ALL k of the reflected grid; we
                # only need the diagonal.  Evaluate per k via the
                # transform of each row and spline interpolation at
                # that k (reference semantics).
                for i, ki in enumerate(khi):
                    kk, G = _mellin_sph_bessel(
                        r, r ** (3 - n) * f[i], n, q=1.5 - n)
                    I = 4 * numpy.pi * G
                    m = len(kk) // 8
                    Pzel[i] += spline(kk[m:-m], I[m:-m])(ki)
            out[hi] = Pzel
        return out[0] if scalar else out
