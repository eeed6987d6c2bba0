"""
Correlation-function <-> power-spectrum transforms (reference
nbodykit/cosmology/correlation.py:8-200).  The reference delegates the
spherical-Bessel Hankel transform to the external ``mcfit`` package
(absent here, unpinned in the reference's requirements); this module
restates the published FFTLog algorithm (Hamilton 2000, MNRAS 312, 257)
it implements:

    xi_l(r) = i^l / (2 pi^2)  int k^2 dk j_l(kr) P_l(k)
    P_l(k)  = 4 pi (-i)^l     int r^2 dr j_l(kr) xi_l(r)

On a log-periodic grid the integral is diagonal in Mellin space with
kernel U_mu(s) = 2^{s-1} Gamma((mu+s)/2) / Gamma((mu-s)/2 + 1) for
J_mu; j_l maps to mu = l + 1/2.  Validated against direct quadrature
and the Gaussian analytic pair in tests/test_correlation_cpu.py.
"""
import numpy
from scipy.interpolate import InterpolatedUnivariateSpline
from scipy.special import loggamma

NUM_PTS = 1024


def _sph_bessel_fftlog(x, a, ell, prefac_exp):
    """Core FFTLog: G(y) = int_0^inf f(x) j_ell(x y) x^2 dx evaluated
    on the reflected log grid y = 1/x (reversed), given
    a = f(x) * x^{3/2} samples on log-even x.  ``prefac_exp`` scales the
    output (the caller's 1/(2 pi^2) or 4 pi).

    Derivation: with j_l(t) = sqrt(pi/(2t)) J_mu(t), mu = l + 1/2,
        G(y) = sqrt(pi/2) y^{-3/2} (1/N) sum_m c_m (x0 y)^{-i eta_m}
                U(1 + i eta_m)
    where c_m are the log-Fourier coefficients of a(x) and
        U(s) = 2^{s-1} Gamma((mu+s)/2) / Gamma((mu-s)/2 + 1).
    """
    N = len(x)
    lnx = numpy.log(x)
    dln = (lnx[-1] - lnx[0]) / (N - 1)
    mu = ell + 0.5

    # log-Fourier coefficients of a; frequencies eta_m = 2 pi m / (N dln)
    c = numpy.fft.rfft(a)
    m = numpy.arange(len(c))
    eta = 2 * numpy.pi * m / (N * dln)

    # U(1 + i eta) via log-gamma for stability
    s = 1.0 + 1j * eta
    lnU = (s - 1) * numpy.log(2.0) + loggamma((mu + s) / 2) \
        - loggamma((mu - s) / 2 + 1)
    U = numpy.exp(lnU)

    # output grid: same spacing, spanning 1/x[-1] .. 1/x[0]
    y0 = 1.0 / x[-1]
    y = y0 * numpy.exp(numpy.arange(N) * dln)

    # G(y_j) = sqrt(pi/2) y_j^{-3/2} (1/N) sum_m c_m (x0 y0)^{-i eta}
    #          U(1+i eta) e^{-2 pi i m j / N};  the e^{-...} sign flips
    # irfft's convention, so the output is read at index (N - j) % N
    # (validated to machine precision on the Gaussian pair)
    phase = (x[0] * y0) ** (-1j * eta)
    g = numpy.fft.irfft(c * U * phase, N)
    idx = (N - numpy.arange(N)) % N
    G = prefac_exp * numpy.sqrt(numpy.pi / 2) * y ** (-1.5) * g[idx]
    return y, G


def _mellin_sph_bessel(x, F, ell, q):
    """Biased FFTLog (mcfit's core): G(y) = int_0^inf F(x) j_ell(xy)
    dx/x on the reflected log grid, with Mellin tilt ``q`` (F is
    decomposed as x^q times a log-periodic part).  Kernel
    M(s) = int t^{s-1} j_ell(t) dt
         = sqrt(pi/2) 2^{s-3/2} Gamma((ell+s)/2) / Gamma((ell+3-s)/2).
    Validated against direct quadrature in tests/test_correlation_cpu.py.
    (Low-ringing phase rotation is not implemented; callers trim the
    grid edges.)"""
    N = len(x)
    dln = (numpy.log(x[-1]) - numpy.log(x[0])) / (N - 1)
    b = F * x ** (-q)
    c = numpy.fft.rfft(b)
    m = numpy.arange(len(c))
    eta = 2 * numpy.pi * m / (N * dln)

    s = q + 1j * eta
    lnM = 0.5 * numpy.log(numpy.pi / 2) + (s - 1.5) * numpy.log(2.0) \
        + loggamma((ell + s) / 2) - loggamma((ell + 3 - s) / 2)
    M = numpy.exp(lnM)

    y0 = 1.0 / x[-1]
    y = y0 * numpy.exp(numpy.arange(N) * dln)
    phase = (x[0] * y0) ** (-1j * eta)
    g = numpy.fft.irfft(c * M * phase, N)
    idx = (N - numpy.arange(N)) % N
    return y, y ** (-q) * g[idx]


def _resample_log(k, F, N=NUM_PTS):
    """Spline onto a log-even grid spanning the input range."""
    k = numpy.asarray(k, dtype='f8')
    F = numpy.asarray(F, dtype='f8')
    spl = InterpolatedUnivariateSpline(numpy.log(k), F)
    lnk = numpy.linspace(numpy.log(k[0]), numpy.log(k[-1]), N)
    return numpy.exp(lnk), spl(lnk)


def pk_to_xi(k, Pk, ell=0, extrap=True, Nfft=NUM_PTS):
    """Spline of xi_ell(r) from sampled P_ell(k) (reference :39-68;
    mcfit.P2xi semantics: the i^ell phase makes even multipoles real,
    with sign (-1)^{ell/2})."""
    if ell % 2:
        raise ValueError("odd multipoles are imaginary in this "
                         "convention; even ell only")
    kk, FF = _resample_log(k, Pk, Nfft)
    sign = (-1.0) ** (ell // 2)
    a = FF * kk ** 1.5
    r, xi = _sph_bessel_fftlog(kk, a, ell,
                               prefac_exp=sign / (2 * numpy.pi ** 2))
    # keep the well-sampled interior (FFTLog edges ring)
    sl = slice(Nfft // 8, -Nfft // 8)
    return InterpolatedUnivariateSpline(r[sl], xi[sl])


def xi_to_pk(r, xi, ell=0, extrap=False, Nfft=NUM_PTS):
    """Spline of P_ell(k) from sampled xi_ell(r) (reference :8-36;
    mcfit.xi2P semantics)."""
    if ell % 2:
        raise ValueError("odd multipoles are imaginary in this "
                         "convention; even ell only")
    rr, FF = _resample_log(r, xi, Nfft)
    sign = (-1.0) ** (ell // 2)
    a = FF * rr ** 1.5
    k, Pk = _sph_bessel_fftlog(rr, a, ell,
                               prefac_exp=sign * 4 * numpy.pi)
    sl = slice(Nfft // 8, -Nfft // 8)
    return InterpolatedUnivariateSpline(k[sl], Pk[sl])


class CorrelationFunction(object):
    """xi(r) of a callable power spectrum (reference :71-140): evaluate
    P on a log grid and FFTLog it; re-evaluates when the power object's
    ``redshift``/``sigma8`` attributes change."""

    def __init__(self, power):
        self.power = power
        self._spline = None
        self._state = None

    def _refresh(self):
        state = (getattr(self.power, 'redshift', None),
                 getattr(self.power, 'sigma8', None))
        if self._spline is None or state != self._state:
            kmin = getattr(self.power, 'kmin', 1e-5)
            kmax = getattr(self.power, 'kmax', 1e2)
            k = numpy.logspace(numpy.log10(kmin), numpy.log10(kmax),
                               NUM_PTS)
            self._spline = pk_to_xi(k, self.power(k))
            self._state = state

    @property
    def redshift(self):
        return self.power.redshift

    @property
    def sigma8(self):
        return self.power.sigma8

    @property
    def attrs(self):
        return getattr(self.power, 'attrs', {})

    def __call__(self, r):
        self._refresh()
        return self._spline(r)
