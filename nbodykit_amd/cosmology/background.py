"""
Matter-dominated linear growth from the single-fluid ODE.

Restates ``nbodykit/cosmology/background.py:4-256`` (MatterDominated):
solve D1''(lna) = -(2 + dlnE/dlna) D1' + 1.5 Om(a) D1 from matter-dominated
initial conditions D1 = a at a = 1e-7, normalize D1(a=1) = 1; the growth
rate is f1 = dlnD1/dlna.  (Second-order 2LPT growth D2 is also integrated —
the ODE carries it for free — though the hot path only uses D1/f1.)
"""
import numpy
from scipy.integrate import odeint


class MatterDominated(object):

    def __init__(self, Omega0_m, Omega0_lambda=None, Omega0_k=0,
                 a=None, a_normalize=1.0):
        if Omega0_lambda is None:
            Omega0_lambda = 1 - Omega0_k - Omega0_m
        self.Omega0_m = Omega0_m
        self.Omega0_lambda = Omega0_lambda
        self.Omega0_k = Omega0_k

        if a is None:
            lna = numpy.log(numpy.logspace(-7, 0, 1024 * 10, endpoint=True))
        else:
            a = numpy.array(a, copy=True).ravel()
            if a_normalize not in a:
                a = numpy.concatenate([[a_normalize], a])
            a.sort()
            if a[0] > 1e-7:
                a = numpy.concatenate([[1e-7], a])
            lna = numpy.log(a)
        self.lna = lna

        self._D1, self._D2 = self._integrate(a_normalize)

    # background functions (flat/curved LCDM, no radiation) ---------------
    def efunc(self, a):
        return (self.Omega0_m / a ** 3 + self.Omega0_k / a ** 2
                + self.Omega0_lambda) ** 0.5

    def efunc_prime(self, a):
        """dE/da."""
        return 0.5 / self.efunc(a) * (-3 * self.Omega0_m / a ** 4
                                      - 2 * self.Omega0_k / a ** 3)

    def Om(self, a):
        return (self.Omega0_m / a ** 3) / self.efunc(a) ** 2

    # growth ODE ----------------------------------------------------------
    def _ode(self, y, lna):
        D1, F1, D2, F2 = y
        a = numpy.exp(lna)
        # hfac = -2 - dlnE/dlna
        hfac = -2.0 - self.efunc_prime(a) * a / self.efunc(a)
        omega = self.Om(a)
        return (F1,
                hfac * F1 + 1.5 * omega * D1,
                F2,
                hfac * F2 + 1.5 * omega * D2 - 1.5 * omega * D1 ** 2)

    def _integrate(self, a_normalize):
        a0 = numpy.exp(self.lna[0])
        # matter-dominated ICs: D1 ~ a, D2 ~ -3/7 a^2
        y0 = [a0, a0, -3.0 / 7 * a0 ** 2, -6.0 / 7 * a0 ** 2]

        y = odeint(self._ode, y0, self.lna, tcrit=[0.], atol=0)

        # store (D, D', D'') per time step; D'' from the ODE itself
        v1 = numpy.empty((len(self.lna), 3))
        v2 = numpy.empty((len(self.lna), 3))
        for i, (yi, lnai) in enumerate(zip(y, self.lna)):
            D1, F1, D2, F2 = yi
            _, F1p, _, F2p = self._ode(yi, lnai)
            v1[i] = (D1, F1, F1p)
            v2[i] = (D2, F2, F2p)

        ind = numpy.abs(self.lna - numpy.log(a_normalize)).argmin()
        v1 /= v1[ind][0]
        v2 /= v2[ind][0]
        return v1, v2

    # interpolated accessors ----------------------------------------------
    def D1(self, a, order=0):
        return numpy.interp(numpy.log(a), self.lna, self._D1[:, order])

    def D2(self, a, order=0):
        return numpy.interp(numpy.log(a), self.lna, self._D2[:, order])

    def f1(self, a):
        return self.D1(a, order=1) / self.D1(a, order=0)

    def f2(self, a):
        return self.D2(a, order=1) / self.D2(a, order=0)
