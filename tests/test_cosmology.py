"""
EH LinearPower + growth slice (reference power/linear.py:33-156,
transfers.py:73-255, background.py:39-256).
"""
import numpy
import pytest
from numpy.testing import assert_allclose

from nbodykit_amd.cosmology import (Cosmology, Planck15, LinearPower,
                                    EisensteinHu, NoWiggleEisensteinHu)


def test_growth_normalization():
    # D1(z=0) == 1, growing with a
    assert_allclose(Planck15.scale_independent_growth_factor(0.), 1.0,
                    rtol=1e-10)
    D = Planck15.scale_independent_growth_factor([0., 0.55, 1.0])
    assert D[0] > D[1] > D[2] > 0


def test_growth_rate_matter_limit():
    # high z: matter domination, f -> 1
    f = Planck15.scale_independent_growth_rate(50.)
    assert_allclose(f, 1.0, rtol=1e-2)
    # z=0 LCDM: f ~ Om(z=0)^0.55 ~ 0.52
    f0 = Planck15.scale_independent_growth_rate(0.)
    assert_allclose(f0, Planck15.Omega0_m ** 0.55, rtol=2e-2)


def test_efunc():
    assert_allclose(Planck15.efunc(0.), 1.0, rtol=1e-12)
    om, ol = Planck15.Omega0_m, 1 - Planck15.Omega0_m
    assert_allclose(Planck15.efunc(1.), numpy.sqrt(om * 8 + ol), rtol=1e-12)


def test_sigma8_normalization():
    P = LinearPower(Planck15, redshift=0., transfer='EisensteinHu')
    assert_allclose(P.sigma_r(8.), Planck15.sigma8, rtol=1e-4)


def test_redshift_scaling():
    P0 = LinearPower(Planck15, redshift=0., transfer='EisensteinHu')
    P1 = LinearPower(Planck15, redshift=0.55, transfer='EisensteinHu')
    D = Planck15.scale_independent_growth_factor(0.55)
    k = numpy.logspace(-3, 0, 10)
    assert_allclose(P1(k), P0(k) * D ** 2, rtol=1e-10)


def test_transfer_normalized_at_low_k():
    for cls in (EisensteinHu, NoWiggleEisensteinHu):
        T = cls(Planck15, 0.)
        assert_allclose(T(numpy.array([1e-7])), 1.0, rtol=1e-3)
        assert T(0.) == 1.0


def test_wiggles_vs_nowiggle():
    Pw = LinearPower(Planck15, 0., 'EisensteinHu')
    Pn = LinearPower(Planck15, 0., 'NoWiggleEisensteinHu')
    k = numpy.logspace(-2, 0, 200)
    ratio = Pw(k) / Pn(k)
    # BAO wiggles oscillate around ~1 at the few-percent level
    assert numpy.abs(ratio - 1).max() < 0.2
    assert numpy.abs(ratio - 1).max() > 0.01


def test_sigma8_setter():
    P = LinearPower(Planck15, 0., 'EisensteinHu')
    base = P(0.1)
    P.sigma8 = Planck15.sigma8 * 2
    assert_allclose(P(0.1), base * 4, rtol=1e-12)


def test_class_transfer_rejected():
    with pytest.raises(ValueError):
        LinearPower(Planck15, 0., transfer='CLASS')


def test_cosmology_clone_and_dict():
    c2 = Planck15.clone(sigma8=0.9)
    assert c2.sigma8 == 0.9 and Planck15.sigma8 == 0.8159
    d = dict(c2.pars)
    assert d['h'] == 0.6774


def test_named_cosmologies_and_distances():
    import numpy
    from numpy.testing import assert_allclose
    from nbodykit_amd.cosmology import (Planck13, Planck15, WMAP5,
                                        WMAP7, WMAP9)
    for c in (Planck13, Planck15, WMAP5, WMAP7, WMAP9):
        assert 0.6 < c.h < 0.75
        # LinearPower works for every named set
        from nbodykit_amd.cosmology import LinearPower
        P = LinearPower(c, redshift=0.0)
        assert P(0.1) > 0
    z = numpy.array([0.5, 1.0])
    dc = Planck15.comoving_distance(z)
    assert_allclose(Planck15.angular_diameter_distance(z), dc / (1 + z))
    assert_allclose(Planck15.luminosity_distance(z), dc * (1 + z))
    assert_allclose(Planck15.comoving_transverse_distance(z), dc)
