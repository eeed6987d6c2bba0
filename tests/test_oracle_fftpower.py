"""
Ports of the reference's FFTPower/CatalogMesh/meshtools property tests to
the oracle (test provenance in each docstring).  These jointly pin the
window kernels + FFT normalization + Hermitian convention + binning
 — the oracle is then the parity anchor for the GPU path.
"""
import numpy
import pytest
from numpy.testing import assert_allclose, assert_array_equal

from oracle import (MeshGeometry, r2c, c2r, complex_coords, paint,
                    to_real_field, fftpower_oracle, apply_compensation)
from tests.conftest import uniform_positions


@pytest.fixture(scope='module')
def upos():
    return uniform_positions(3e-4, 512., seed=42)


def _flat_chi2(r):
    Pk = r['power'].real
    sel = ~numpy.isnan(Pk)
    err = (2 * Pk[sel] ** 2 / r['modes'][sel]) ** 0.5
    residual = (Pk[sel] - r['attrs']['shotnoise']) / err
    return (residual ** 2).sum() / sel.sum()


def test_cic_aliasing(upos):
    """compensated CIC of uniform -> flat shot noise, chi2 < 1
    (algorithms/tests/test_fftpower.py:29-44)"""
    r = fftpower_oracle(upos, Nmesh=64, BoxSize=512., mode='1d',
                        resampler='cic', compensated=True, kmin=0.02)
    assert _flat_chi2(r) < 1.0


def test_tsc_aliasing(upos):
    """(test_fftpower.py:12-26)"""
    r = fftpower_oracle(upos, Nmesh=64, BoxSize=512., mode='1d',
                        resampler='tsc', compensated=True, kmin=0.02)
    assert _flat_chi2(r) < 1.0


def test_pcs_aliasing(upos):
    """same property for the PCS window (support 4)"""
    r = fftpower_oracle(upos, Nmesh=64, BoxSize=512., mode='1d',
                        resampler='pcs', compensated=True, kmin=0.02)
    assert _flat_chi2(r) < 1.0


def test_tsc_interlacing(upos):
    """interlaced+compensated TSC -> P ~ 1/nbar to 10 %
    (source/mesh/tests/test_catalogmesh.py:12-23)"""
    r = fftpower_oracle(upos, Nmesh=64, BoxSize=512., mode='1d',
                        resampler='tsc', compensated=True, interlaced=True,
                        kmin=0.02)
    P = r['power'].real[5:]
    assert_allclose(P[~numpy.isnan(P)], 1 / 3e-4, rtol=1e-1)


def test_cic_interlacing(upos):
    """(test_catalogmesh.py:85-99)"""
    r = fftpower_oracle(upos, Nmesh=64, BoxSize=512., mode='1d',
                        resampler='cic', compensated=True, interlaced=True,
                        kmin=0.02)
    P = r['power'].real[5:]
    assert_allclose(P[~numpy.isnan(P)], 1 / 3e-4, rtol=1e-1)


def test_fftpower_poles():
    """P(k,mu)-weighted monopole == P_0 exactly (test_fftpower.py:49-61)"""
    pos = uniform_positions(3e-3, 512., seed=42)
    r = fftpower_oracle(pos, Nmesh=32, BoxSize=1024., mode='2d',
                        poles=[0, 2, 4])
    pkmu = r['power'].real
    modes = r['modes']
    modes_1d = modes.sum(axis=-1)
    mono_from_pkmu = numpy.nansum(pkmu * modes, axis=-1) / modes_1d
    assert_array_equal(modes_1d, r['pole_modes'])
    assert_allclose(mono_from_pkmu, r['poles'][0].real)


def test_fftpower_unique(upos):
    """dk=0 -> unique-modulus bins whose means equal the edges' centers
    (test_fftpower.py:65-71)"""
    r = fftpower_oracle(upos, Nmesh=32, BoxSize=512., mode='1d', dk=0)
    assert_allclose(r['kcoords'], r['k'], rtol=1e-6)


def test_fftpower_zero_mode(upos):
    """zero mode cleared but binned (test_fftpower.py:101-107 and the
    parity trap in SURVEY §8a)"""
    r = fftpower_oracle(upos, Nmesh=32, BoxSize=512., mode='1d')
    assert_array_equal(r['power'][0], 0)
    # the k=0 mode is still counted in the first bin (kmin=0 default)
    assert r['modes'][0] >= 1


def test_paint_empty():
    """empty catalog -> normalized field 1.0, raw field 0.0
    (test_catalogmesh.py:27-43)"""
    geom = MeshGeometry(64, 512.)
    empty = numpy.empty((0, 3))
    with numpy.errstate(invalid='ignore', divide='ignore'):
        mesh, attrs = to_real_field(empty, geom, resampler='tsc',
                                    interlaced=True, normalize=True)
        assert_allclose(mesh, 1.0)
        mesh, attrs = to_real_field(empty, geom, resampler='tsc',
                                    interlaced=True, normalize=False)
        assert_allclose(mesh, 0.0)


def test_paint_chunksize(upos):
    """result independent of paint_chunk_size (test_catalogmesh.py:47-60)"""
    geom = MeshGeometry(64, 512.)
    m1, _ = to_real_field(upos, geom, resampler='tsc', interlaced=True,
                          paint_chunk_size=len(upos) // 4)
    m2, _ = to_real_field(upos, geom, resampler='tsc', interlaced=True,
                          paint_chunk_size=len(upos))
    assert_allclose(m1, m2)


def test_shotnoise_weighted(upos):
    """weighted shot noise V*W2/W^2: uniform weights in [0,1) give
    SN ~ 4/(3 nbar) (test_catalogmesh.py:64-82)"""
    from nbodykit_amd.comm import SerialComm
    from nbodykit_amd.mpirng import MPIRandomState
    rng = MPIRandomState(SerialComm(), seed=42, size=len(upos))
    # skip the position draws so the weight stream differs
    rng.uniform(itemshape=(3,))
    rng.uniform(itemshape=(3,))
    w = rng.uniform()
    geom = MeshGeometry(64, 512.)
    _, attrs = to_real_field(upos, geom, weight=w, resampler='tsc',
                             interlaced=True)
    assert_allclose(attrs['shotnoise'], 4 / 3.0 / 3e-4, rtol=1e-2)


def test_painted_mean_is_one(upos):
    """normalized paint has mean exactly 1+delta with cmean 1
    (source/catalog/tests/test_lognormal.py:14-36 property)"""
    geom = MeshGeometry(32, 512.)
    mesh, attrs = to_real_field(upos, geom, resampler='cic')
    assert_allclose(mesh.mean(), 1.0, rtol=1e-12)


def test_r2c_roundtrip_and_norm():
    """r2c normalized by 1/N^3 (k=0 mode == configuration mean), c2r
    inverse (fftpower.py:126-128, mockmaker.py:27-36)"""
    rng = numpy.random.RandomState(7)
    geom = MeshGeometry(16, 100.)
    field = rng.normal(size=(16, 16, 16))
    c = r2c(field, geom)
    assert_allclose(c[0, 0, 0].real, field.mean(), rtol=1e-12)
    back = c2r(c, geom)
    assert_allclose(back, field, atol=1e-12)


def test_hermitian_weights_convention():
    """weight == 2 iff compressed-axis frequency > 0; Nyquist stored
    negative (nbodykit/tests/test_meshtools.py:67-93,
    meshtools.py:150-153)"""
    geom = MeshGeometry(8, 8.)
    x = complex_coords(geom)
    # compressed axis: 0, 1, 2, 3, then the Nyquist stored negative
    k0 = 2 * numpy.pi / 8.
    assert_allclose(x[2].ravel() / k0, [0, 1, 2, 3, -4])
    # full axes carry fftfreq order
    assert_allclose(x[0].ravel() / k0, [0, 1, 2, 3, -4, -3, -2, -1])


def test_parseval():
    """sum |delta_k|^2 (with Hermitian double-count) == <delta^2>/N^3 —
    ties the compressed layout + normalization together"""
    rng = numpy.random.RandomState(3)
    geom = MeshGeometry(16, 50.)
    field = rng.normal(size=(16, 16, 16))
    c = r2c(field, geom)
    w = numpy.ones(c.shape)
    w[:, :, 1:-1] = 2.0   # double all but DC and Nyquist planes
    total = (numpy.abs(c) ** 2 * w).sum()
    assert_allclose(total, (field ** 2).mean() / field.size * field.size,
                    rtol=1e-10)


def test_compensation_unity_at_k0():
    geom = MeshGeometry(8, 8.)
    c = numpy.ones(geom.cshape, dtype='c16')
    for resampler in ('cic', 'tsc', 'pcs'):
        for interlaced in (False, True):
            cc = c.copy()
            apply_compensation(cc, geom, resampler, interlaced)
            assert_allclose(cc[0, 0, 0], 1.0)
            assert numpy.isfinite(cc).all()
            # compensation amplifies (divides by |W|<=1)
            assert (numpy.abs(cc) >= 1.0 - 1e-12).all()


def test_cross_power_matches_auto(upos):
    """cross power of a catalog with itself == auto power (modulo the
    shot-noise attr, which is zero for cross: fftpower.py:135-140)"""
    r_auto = fftpower_oracle(upos, Nmesh=32, BoxSize=512., mode='1d')
    r_cross = fftpower_oracle(upos, Nmesh=32, BoxSize=512., mode='1d',
                              second_position=upos)
    assert_allclose(r_auto['power'], r_cross['power'], rtol=1e-13)
    assert r_cross['attrs']['shotnoise'] == 0.0


def test_mode_validation(upos):
    with pytest.raises(ValueError):
        fftpower_oracle(upos, Nmesh=8, BoxSize=512., mode='3d')


# ---- FFTCorr oracle (reference algorithms/fftcorr.py) ------------------

def test_fftcorr_shotnoise_spike(upos):
    """xi of a Poisson catalog: the r=0 bin holds ~SN/Vcell (the c2r of
    the flat shot-noise spectrum), everything else ~0"""
    from oracle import fftcorr_oracle
    r = fftcorr_oracle(upos, Nmesh=32, BoxSize=512., mode='1d')
    xi = r['corr']
    Vcell = (512. / 32) ** 3
    expect = r['attrs']['shotnoise'] / Vcell
    assert abs(xi[0] / expect - 1) < 0.05
    tail = xi[3:][~numpy.isnan(xi[3:])]
    assert numpy.sqrt((tail ** 2).mean()) < 0.05 * xi[0]


def test_fftcorr_poles_identity(upos):
    """mu-weighted monopole == xi_0 (the FFTPower identity in r-space)"""
    from oracle import fftcorr_oracle
    r = fftcorr_oracle(upos, Nmesh=32, BoxSize=512., mode='2d', Nmu=4,
                       poles=[0, 2])
    ximu = r['corr']
    modes = r['modes']
    modes_1d = modes.sum(axis=-1)
    mono = numpy.nansum(ximu * modes, axis=-1) / modes_1d
    assert_array_equal(modes_1d, r['pole_modes'])
    ok = numpy.isfinite(mono)
    assert_allclose(mono[ok], r['poles'][0][ok], rtol=1e-10, atol=1e-12)


def test_fftcorr_mode_validation(upos):
    from oracle import fftcorr_oracle
    with pytest.raises(ValueError):
        fftcorr_oracle(upos, Nmesh=8, BoxSize=512., mode='3d')


# ---- readout + FFTRecon oracle (reference fftrecon.py) ------------------

def test_readout_at_grid_points():
    """CIC/nnb readout exactly at grid points returns the mesh values
    (TSC/PCS smooth over neighbours even there); every window is a
    partition of unity, so a constant field reads back constant
    anywhere."""
    from oracle import readout
    geom = MeshGeometry(8, 16.)
    rng = numpy.random.RandomState(5)
    mesh = rng.normal(size=(8, 8, 8))
    ii = numpy.stack(numpy.meshgrid(*[numpy.arange(8)] * 3,
                                    indexing='ij'), axis=-1).reshape(-1, 3)
    pos = ii * (16. / 8)
    for resampler in ('cic', 'nnb'):
        vals = readout(pos, mesh, geom, resampler=resampler)
        assert_allclose(vals, mesh.reshape(-1), rtol=1e-12, atol=1e-12)
    anywhere = rng.uniform(0, 16., size=(200, 3))
    const = numpy.full((8, 8, 8), 3.25)
    for resampler in ('cic', 'tsc', 'pcs', 'nnb'):
        vals = readout(anywhere, const, geom, resampler=resampler)
        assert_allclose(vals, 3.25, rtol=1e-12)


def test_readout_paint_adjoint():
    """<paint(pos, m), mesh> == <m, readout(pos, mesh)> — paint and
    readout are adjoint (same window weights)"""
    from oracle import readout
    geom = MeshGeometry(16, 32.)
    rng = numpy.random.RandomState(6)
    pos = rng.uniform(0, 32., size=(500, 3))
    m = rng.uniform(0.5, 1.5, size=500)
    mesh = rng.normal(size=(16, 16, 16))
    for resampler in ('cic', 'tsc', 'pcs'):
        painted = numpy.zeros((16, 16, 16))
        paint(pos, m, painted, geom, resampler=resampler)
        lhs = (painted * mesh).sum()
        rhs = (m * readout(pos, mesh, geom, resampler=resampler)).sum()
        assert_allclose(lhs, rhs, rtol=1e-12)


def test_fftrecon_data_equals_randoms_is_null():
    """LGS with data == randoms: identical shifts => identically zero"""
    from oracle import fftrecon_oracle
    pos = uniform_positions(2e-3, 64., seed=9)
    out = fftrecon_oracle(pos, pos, Nmesh=16, BoxSize=64., bias=1.0,
                          f=0.0, R=10.0, scheme='LGS')
    assert_allclose(out, 0.0, atol=1e-12)


def test_fftrecon_schemes_run_and_mean_zero():
    from oracle import fftrecon_oracle
    data = uniform_positions(2e-3, 64., seed=9)
    ran = uniform_positions(4e-3, 64., seed=10)
    for scheme in ('LGS', 'LF2', 'LRR'):
        out = fftrecon_oracle(data, ran, Nmesh=16, BoxSize=64., bias=1.2,
                              f=0.3, R=10.0, scheme=scheme)
        # delta fields: mean ~ 0 (data mean 1 minus randoms mean 1)
        assert abs(out.mean()) < 1e-10
        assert numpy.isfinite(out).all()
