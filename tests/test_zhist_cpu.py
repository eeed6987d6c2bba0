"""CPU tests for RedshiftHistogram (reference algorithms/zhist.py):
shell-volume-normalized n(z), Scott's rule, interpolation, save/load."""
import numpy
import numpy.testing as nt
import pytest

from nbodykit_amd.lab import ArrayCatalog, RedshiftHistogram
from nbodykit_amd.cosmology import Planck15


def _cat(n=20000, zlo=0.2, zhi=0.8, seed=0):
    rng = numpy.random.RandomState(seed)
    return ArrayCatalog({'Redshift': rng.uniform(zlo, zhi, n),
                         'W': rng.exponential(size=n)})


def test_nbar_recovers_constant_density():
    # objects drawn with constant comoving density: nbar(z) ~ const
    rng = numpy.random.RandomState(1)
    n = 200000
    fsky = 0.25
    zgrid = numpy.linspace(0.2, 0.8, 512)
    r = Planck15.comoving_distance(zgrid)
    V = (4. / 3) * numpy.pi * (r ** 3 - r[0] ** 3) * fsky
    # sample z with dN/dz ~ dV/dz by inverse-CDF on the volume
    u = rng.uniform(0, V[-1], n)
    z = numpy.interp(u, V, zgrid)
    cat = ArrayCatalog({'Redshift': z})
    h = RedshiftHistogram(cat, fsky, Planck15, bins=20)
    target = n / V[-1]
    nt.assert_allclose(h.nbar, target, rtol=0.1)
    # interpolation ~ histogram at centers; zeros outside by default
    nt.assert_allclose(h.interpolate(h.bin_centers), h.nbar, rtol=1e-10)
    assert h.interpolate(numpy.array([5.0]))[0] == 0.0


def test_bins_and_weights():
    cat = _cat()
    h = RedshiftHistogram(cat, 0.5, Planck15, bins=10)
    assert len(h.bin_edges) == 11
    # total weighted counts conserved over the right-open bin range
    # (the reference's searchsorted(..., 'right') drops z == max edge)
    hw = RedshiftHistogram(cat, 0.5, Planck15, bins=10, weight='W')
    z = numpy.asarray(cat['Redshift'])
    w = numpy.asarray(cat['W'])
    inside = (z >= hw.bin_edges[0]) & (z < hw.bin_edges[-1])
    nt.assert_allclose((hw.nbar * hw.dV).sum(), w[inside].sum(),
                       rtol=1e-10)
    # Scott's rule picks a sane bin count
    hs = RedshiftHistogram(cat, 0.5, Planck15)
    assert 5 < len(hs.bin_edges) < 500


def test_missing_column():
    cat = _cat()
    with pytest.raises(ValueError):
        RedshiftHistogram(cat, 0.5, Planck15, redshift='zz')


def test_save_load(tmp_path):
    cat = _cat(2000)
    h = RedshiftHistogram(cat, 0.5, Planck15, bins=8)
    fn = str(tmp_path / 'nz.json')
    h.save(fn)
    h2 = RedshiftHistogram.load(fn)
    nt.assert_allclose(h2.nbar, h.nbar)
    nt.assert_allclose(h2.dV, h.dV)
