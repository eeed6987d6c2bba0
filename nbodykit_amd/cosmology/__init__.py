"""
Minimal cosmology slice for the FFTPower hot path:
Eisenstein-Hu transfer + sigma8-normalized LinearPower + matter-dominated
growth from the ODE.  The reference's full Boltzmann wrapper (classylss /
CLASS, nbodykit/cosmology/cosmology.py) is out of scope; only the pieces
LogNormalCatalog and the bench configs need are built:

- :class:`Cosmology`: parameter container (h, Omega0_m, Omega0_b, n_s,
  Tcmb0, sigma8) with ``efunc`` / ``scale_independent_growth_factor`` /
  ``scale_independent_growth_rate`` backed by :class:`MatterDominated`
  (reference nbodykit/cosmology/background.py:39-104, 207-256).
- :class:`EisensteinHu` / :class:`NoWiggleEisensteinHu` transfers
  (reference nbodykit/cosmology/power/transfers.py:73-255).
- :class:`LinearPower` (reference nbodykit/cosmology/power/linear.py:5-156);
  ``sigma_r`` uses direct log-k quadrature instead of mcfit's FFTLog
  (mcfit is unavailable; the integrand is smooth, and a dense trapezoid on
  4096 log-spaced points agrees with FFTLog far below the 1e-5 parity bar).
"""
import numpy

from .background import MatterDominated
from .transfers import EisensteinHu, NoWiggleEisensteinHu

available_transfers = ['EisensteinHu', 'NoWiggleEisensteinHu']


class Cosmology(object):
    """
    Flat-LCDM parameter container sufficient for the EH transfer and
    matter-dominated growth.  ``dict(cosmo)`` yields the parameters
    (the reference stores ``dict(cosmo)`` in attrs,
    source/catalog/lognormal.py:76).
    """

    def __init__(self, h=0.6774, Omega0_m=0.3089, Omega0_b=0.0486,
                 n_s=0.9667, Tcmb0=2.7255, sigma8=0.8159, **extra):
        self.pars = dict(h=h, Omega0_m=Omega0_m, Omega0_b=Omega0_b,
                         n_s=n_s, Tcmb0=Tcmb0, sigma8=sigma8)
        self.pars.update(extra)
        self._growth = None

    # parameter access ----------------------------------------------------
    @property
    def h(self): return self.pars['h']

    @property
    def Omega0_m(self): return self.pars['Omega0_m']

    @property
    def Omega0_b(self): return self.pars['Omega0_b']

    @property
    def n_s(self): return self.pars['n_s']

    @property
    def Tcmb0(self): return self.pars['Tcmb0']

    @property
    def sigma8(self): return self.pars['sigma8']

    def keys(self):
        return self.pars.keys()

    def __getitem__(self, key):
        return self.pars[key]

    def __iter__(self):
        return iter(self.pars)

    def clone(self, **changes):
        pars = dict(self.pars)
        pars.update(changes)
        return Cosmology(**pars)

    def __eq__(self, other):
        return isinstance(other, Cosmology) and self.pars == other.pars

    # background ----------------------------------------------------------
    @property
    def growth(self):
        if self._growth is None:
            self._growth = MatterDominated(Omega0_m=self.Omega0_m)
        return self._growth

    def efunc(self, z):
        """E(z) = H(z)/H0 for flat LCDM
        (reference background.py:249-250, with a = 1/(1+z))."""
        a = 1.0 / (1.0 + numpy.asarray(z, dtype='f8'))
        return self.growth.efunc(a)

    def scale_independent_growth_factor(self, z):
        """D1(z), normalized to 1 at z=0 (reference background.py:39-54)."""
        a = 1.0 / (1.0 + numpy.asarray(z, dtype='f8'))
        return self.growth.D1(a)

    def scale_independent_growth_rate(self, z):
        """f1(z) = dlnD1/dlna (reference background.py:73-87)."""
        a = 1.0 / (1.0 + numpy.asarray(z, dtype='f8'))
        return self.growth.f1(a)

    def comoving_distance(self, z):
        """Line-of-sight comoving distance in Mpc/h:
        Dc = (c/H0) int_0^z dz'/E(z') (the reference wraps CLASS's
        background; flat matter+Lambda integral here — the neglected
        radiation term shifts Dc by <0.1% at survey redshifts).  The
        integral is tabulated once on a fixed log grid so repeated
        calls with different ranges agree to interpolation accuracy
        (transform.CartesianToSky inverts one call against another)."""
        z = numpy.asarray(z, dtype='f8')
        zmax = float(z.max()) if z.size else 0.0
        cached = getattr(self, '_dc_table', None)
        if cached is None or cached[0] < zmax:
            ztop = max(10.0, 2.0 * zmax)
            grid = numpy.concatenate(
                [[0.0], numpy.logspace(-8, numpy.log10(ztop), 8192)])
            integrand = 1.0 / self.efunc(grid)
            dc = numpy.concatenate([[0.0], numpy.cumsum(
                0.5 * (integrand[1:] + integrand[:-1])
                * numpy.diff(grid))])
            self._dc_table = (ztop, grid, dc)
            cached = self._dc_table
        # c / H0 with H0 = 100 h km/s/Mpc -> Mpc/h units
        return 2997.92458 * numpy.interp(z, cached[1], cached[2])

    def comoving_transverse_distance(self, z):
        """Flat universe: equals the comoving distance (Mpc/h)."""
        return self.comoving_distance(z)

    def angular_diameter_distance(self, z):
        """D_A = D_C / (1+z) for flat LCDM (Mpc/h)."""
        z = numpy.asarray(z, dtype='f8')
        return self.comoving_distance(z) / (1.0 + z)

    def luminosity_distance(self, z):
        """D_L = (1+z) D_C for flat LCDM (Mpc/h)."""
        z = numpy.asarray(z, dtype='f8')
        return self.comoving_distance(z) * (1.0 + z)


from .correlation import (CorrelationFunction, pk_to_xi,   # noqa: E402
                          xi_to_pk)
from .zeldovich import ZeldovichPower                       # noqa: E402

# Named cosmologies: astropy's FlatLambdaCDM parameter sets + the
# sigma8/n_s values nbodykit adds (reference cosmology/__init__.py:8-50;
# the reference builds them via Cosmology.from_astropy over CLASS).
Planck15 = Cosmology(h=0.6774, Omega0_m=0.3089, Omega0_b=0.0486,
                     n_s=0.9667, Tcmb0=2.7255, sigma8=0.8159)

Planck13 = Cosmology(h=0.6777, Omega0_m=0.30712, Omega0_b=0.048252,
                     n_s=0.9611, Tcmb0=2.7255, sigma8=0.8288)

WMAP5 = Cosmology(h=0.702, Omega0_m=0.277, Omega0_b=0.0459,
                  n_s=0.962, Tcmb0=2.725, sigma8=0.817)

WMAP7 = Cosmology(h=0.704, Omega0_m=0.272, Omega0_b=0.0455,
                  n_s=0.967, Tcmb0=2.725, sigma8=0.810)

WMAP9 = Cosmology(h=0.6932, Omega0_m=0.2865, Omega0_b=0.04628,
                  n_s=0.9608, Tcmb0=2.725, sigma8=0.820)


class LinearPower(object):
    """
    sigma8-normalized linear power spectrum
    P(k) = norm * k^n_s * T(k)^2 (reference power/linear.py:114-156).
    ``transfer`` must be 'EisensteinHu' or 'NoWiggleEisensteinHu'
    ('CLASS' needs a Boltzmann code — out of scope here).
    """

    def __init__(self, cosmo, redshift, transfer='EisensteinHu'):
        if transfer not in available_transfers:
            raise ValueError("'transfer' should be one of %s (the CLASS "
                             "transfer needs classylss, which is out of "
                             "scope of this rebuild)" % available_transfers)
        self.cosmo = cosmo.clone()
        self.transfer = transfer
        self._sigma8 = self.cosmo.sigma8

        cls = {'EisensteinHu': EisensteinHu,
               'NoWiggleEisensteinHu': NoWiggleEisensteinHu}[transfer]
        self._transfer = cls(self.cosmo, redshift)

        # normalize at z=0 so that sigma_r(8) == sigma8 (linear.py:57-60)
        self._norm = 1.0
        self.redshift = 0
        self._norm = (self._sigma8 / self.sigma_r(8.)) ** 2

        self.redshift = redshift

        self._attrs = {}
        self._attrs['transfer'] = transfer
        self._attrs['cosmo'] = dict(cosmo.pars)

    @property
    def attrs(self):
        self._attrs['redshift'] = self.redshift
        self._attrs['sigma8'] = self.sigma8
        return self._attrs

    @property
    def redshift(self):
        return self._z

    @redshift.setter
    def redshift(self, value):
        self._z = value
        self._transfer.redshift = value

    @property
    def sigma8(self):
        return self._sigma8

    @sigma8.setter
    def sigma8(self, value):
        self._norm *= (value / self._sigma8) ** 2
        self._sigma8 = value

    def __call__(self, k):
        """P(k) in (Mpc/h)^3 at ``self.redshift``; k in h/Mpc."""
        Pk = numpy.asarray(k) ** self.cosmo.n_s * self._transfer(k) ** 2
        return self._norm * Pk

    def velocity_dispersion(self, kmin=1e-5, kmax=10., **kwargs):
        r"""sigma_v in Mpc/h: sigma_v^2 = 1/(6 pi^2) \int dk P(k)
        (reference linear.py:158-183)."""
        from scipy.integrate import quad

        def integrand(logq):
            q = numpy.exp(logq)
            return q * self(q)
        sigmasq = quad(integrand, numpy.log(kmin), numpy.log(kmax),
                       **kwargs)[0] / (6 * numpy.pi ** 2)
        return sigmasq ** 0.5

    def sigma_r(self, r, kmin=1e-5, kmax=1e1):
        r"""
        RMS mass fluctuation in a top-hat of radius ``r`` Mpc/h:
        sigma^2 = \int dlnk k^3 P(k) / (2 pi^2) W_T(kr)^2 with
        W_T(x) = 3/x^3 (sin x - x cos x)  (reference linear.py:184-218).
        Direct trapezoid over log k (mcfit unavailable; see module docstring).
        """
        k = numpy.logspace(numpy.log10(kmin), numpy.log10(kmax), 4096)
        Pk = self(k)
        scalar = numpy.isscalar(r)
        r = numpy.atleast_1d(numpy.asarray(r, dtype='f8'))
        x = numpy.outer(r, k)
        W = 3.0 / x ** 3 * (numpy.sin(x) - x * numpy.cos(x))
        integrand = k ** 3 * Pk / (2 * numpy.pi ** 2) * W ** 2
        sigmasq = numpy.trapezoid(integrand, numpy.log(k), axis=-1)
        sigma = sigmasq ** 0.5
        return sigma[0] if scalar else sigma
