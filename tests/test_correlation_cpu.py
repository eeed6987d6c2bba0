"""CPU tests for the FFTLog correlation transforms (reference
cosmology/correlation.py; algorithm restated from mcfit/Hamilton 2000 —
the dependency is absent here, so parity anchors are the analytic
Gaussian pair, direct quadrature, and the round trip)."""
import numpy
import numpy.testing as nt
import pytest

from nbodykit_amd.cosmology import (CorrelationFunction, LinearPower,
                                    Planck15, pk_to_xi, xi_to_pk)


def test_gaussian_pair():
    # P = exp(-k^2/2)  <->  xi = (2 pi)^{-3/2} exp(-r^2/2)
    k = numpy.logspace(-4, 3, 700)
    xi = pk_to_xi(k, numpy.exp(-k ** 2 / 2))
    r = numpy.array([0.1, 0.5, 1.0, 2.0, 3.0])
    want = (2 * numpy.pi) ** -1.5 * numpy.exp(-r ** 2 / 2)
    nt.assert_allclose(xi(r), want, rtol=1e-5)


def test_roundtrip():
    k = numpy.logspace(-4, 3, 700)
    P = numpy.exp(-k ** 2 / 2)
    rg = numpy.logspace(-3, 3, 800)
    Pk2 = xi_to_pk(rg, pk_to_xi(k, P)(rg))
    kv = numpy.array([0.1, 0.5, 1.0, 2.0])
    nt.assert_allclose(Pk2(kv), numpy.exp(-kv ** 2 / 2), rtol=1e-4)


def test_quadrupole_against_quadrature():
    from scipy.special import spherical_jn
    k = numpy.logspace(-4, 3, 700)
    P2 = k ** 2 * numpy.exp(-k ** 2 / 2)
    xi2 = pk_to_xi(k, P2, ell=2)
    kk = numpy.linspace(1e-6, 30, 200000)
    for rv in [0.5, 1.0, 2.0]:
        direct = -numpy.trapezoid(
            kk ** 2 * (kk ** 2 * numpy.exp(-kk ** 2 / 2))
            * spherical_jn(2, kk * rv), kk) / (2 * numpy.pi ** 2)
        nt.assert_allclose(xi2(rv), direct, rtol=1e-5)
    with pytest.raises(ValueError):
        pk_to_xi(k, P2, ell=1)


def test_correlation_function_bao():
    # the linear CF of Planck15 has the BAO peak near r ~ 105 Mpc/h
    P = LinearPower(Planck15, redshift=0.0, transfer='EisensteinHu')
    CF = CorrelationFunction(P)
    r = numpy.linspace(60., 140., 161)
    xi = CF(r) * r ** 2
    rpk = r[numpy.argmax(xi)]
    assert 95. < rpk < 115., rpk
    # positive at small separations, decaying
    assert CF(10.) > CF(50.) > 0
