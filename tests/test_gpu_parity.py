"""
End-to-end GPU parity: the product FFTPower (through the public drop-in
API) against the CPU oracle on identical catalogs — the north-star bar:
P(k) within 1e-5 relative per k-bin (f64), mode counts exact — plus the
reference's own full-size property tests run on the GPU.
"""
import glob
import json
import os

import numpy
import pytest
from numpy.testing import assert_allclose, assert_array_equal

pytestmark = pytest.mark.gpu

torch = pytest.importorskip('torch')
if not torch.cuda.is_available():
    pytest.skip('no GPU', allow_module_level=True)

from nbodykit_amd.lab import (UniformCatalog, ArrayCatalog,       # noqa
                              LogNormalCatalog, FFTPower, FieldMesh,
                              LinearPower, Planck15)
from nbodykit_amd import set_options                              # noqa
from nbodykit_amd.utils import JSONDecoder                        # noqa
from oracle import fftpower_oracle                                # noqa
from tests.conftest import uniform_positions                      # noqa

HERE = os.path.dirname(os.path.abspath(__file__))

PARITY_RTOL = 1e-5   # the north-star bar (f64)


def check_parity(r, want, poles=()):
    """product FFTPower result vs oracle dict"""
    assert_array_equal(r.power['modes'], want['modes'])
    got = r.power['power']
    ref = want['power']
    ok = numpy.isfinite(ref.real) & (numpy.abs(ref) > 0)
    rel = numpy.abs(got[ok] - ref[ok]) / numpy.abs(ref[ok])
    assert rel.max() < PARITY_RTOL, 'P parity: %g' % rel.max()
    assert_allclose(numpy.nan_to_num(r.power['k']),
                    numpy.nan_to_num(want['k']), rtol=1e-10, atol=1e-12)
    if 'mu' in r.power.variables:
        assert_allclose(numpy.nan_to_num(r.power['mu']),
                        numpy.nan_to_num(want['mu']),
                        rtol=1e-10, atol=1e-12)
    for ell in poles:
        got = r.poles['power_%d' % ell]
        ref = want['poles'][ell]
        ok = numpy.isfinite(ref.real) & (numpy.abs(ref) > 0)
        rel = numpy.abs(got[ok] - ref[ok]) / numpy.abs(ref[ok])
        assert rel.max() < PARITY_RTOL, 'P_%d parity: %g' % (ell, rel.max())
    assert_allclose(r.attrs['shotnoise'], want['attrs']['shotnoise'],
                    rtol=1e-12)


# ---- parity vs committed golden fixtures (the oracle itself is pinned
# on CPU by tests/test_golden.py; here the GPU path must match them) -----

GOLDEN = sorted(glob.glob(os.path.join(HERE, 'golden',
                                       'oracle_fftpower_*.json')))


@pytest.mark.parametrize('path', GOLDEN,
                         ids=[os.path.basename(p) for p in GOLDEN])
def test_product_matches_golden(path):
    with open(path) as ff:
        g = json.load(ff, cls=JSONDecoder)
    cfg = g['config']
    run = dict(cfg['run'])

    def make_catalog(spec):
        kind, kw = spec
        if kind == 'uniform':
            return UniformCatalog(nbar=kw['nbar'], BoxSize=kw['BoxSize'],
                                  seed=kw['seed'])
        Plin = LinearPower(Planck15, redshift=kw.get('redshift', 0.55))
        return LogNormalCatalog(Plin=Plin, nbar=kw['nbar'],
                                BoxSize=kw['BoxSize'], Nmesh=kw['Nmesh'],
                                bias=kw.get('bias', 2.0), seed=kw['seed'])

    first = make_catalog(cfg['pos'])
    second = make_catalog(cfg['second']) if cfg.get('second') else None

    mesh_kw = dict(Nmesh=run['Nmesh'], resampler=run.get('resampler', 'cic'),
                   compensated=run.get('compensated', True),
                   interlaced=run.get('interlaced', False), dtype='f8')
    m1 = first.to_mesh(**mesh_kw)
    m2 = second.to_mesh(**mesh_kw) if second is not None else None

    r = FFTPower(m1, mode=run['mode'], second=m2,
                 Nmu=run.get('Nmu', 5), kmin=run.get('kmin', 0.),
                 los=run.get('los', [0, 0, 1]),
                 poles=run.get('poles', []))

    assert_array_equal(r.power['modes'], g['modes'])
    got = numpy.ravel(r.power['power'])
    ref = numpy.ravel(g['power'])
    ok = numpy.isfinite(ref.real) & (numpy.abs(ref) > 0)
    rel = numpy.abs(got[ok] - ref[ok]) / numpy.abs(ref[ok])
    assert rel.max() < PARITY_RTOL, 'P parity vs golden: %g' % rel.max()
    for key in [k for k in g if k.startswith('power_')]:
        ell = int(key.split('_')[1])
        got = r.poles['power_%d' % ell]
        ref = g[key]
        ok = numpy.isfinite(ref.real) & (numpy.abs(ref) > 0)
        rel = numpy.abs(got[ok] - ref[ok]) / numpy.abs(ref[ok])
        assert rel.max() < PARITY_RTOL


# ---- live oracle comparisons on shared inputs --------------------------

def test_catalog_parity_cic_1d():
    cat = UniformCatalog(nbar=3e-4, BoxSize=512., seed=42)
    r = FFTPower(cat, mode='1d', Nmesh=64, kmin=0.02)
    pos = uniform_positions(3e-4, 512., 42)
    want = fftpower_oracle(pos, Nmesh=64, BoxSize=512., mode='1d',
                           resampler='cic', compensated=True, kmin=0.02)
    check_parity(r, want)


def test_catalog_parity_tsc_interlaced_2d_poles():
    cat = UniformCatalog(nbar=3e-4, BoxSize=512., seed=42)
    mesh = cat.to_mesh(Nmesh=64, resampler='tsc', compensated=True,
                       interlaced=True, dtype='f8')
    r = FFTPower(mesh, mode='2d', Nmu=5, poles=[0, 2, 4])
    pos = uniform_positions(3e-4, 512., 42)
    want = fftpower_oracle(pos, Nmesh=64, BoxSize=512., mode='2d', Nmu=5,
                           poles=[0, 2, 4], resampler='tsc',
                           compensated=True, interlaced=True)
    check_parity(r, want, poles=[0, 2, 4])


def test_catalog_parity_weighted():
    cat = UniformCatalog(nbar=3e-4, BoxSize=256., seed=11)
    w = numpy.random.RandomState(3).uniform(0.5, 2.0, size=cat.size)
    cat['Weight'] = w
    r = FFTPower(cat, mode='1d', Nmesh=32)
    pos = numpy.asarray(cat['Position'], dtype='f8')
    want = fftpower_oracle(pos, weight=w, Nmesh=32, BoxSize=256.,
                           mode='1d', resampler='cic', compensated=True)
    check_parity(r, want)


def test_cross_power_parity():
    a = UniformCatalog(nbar=3e-4, BoxSize=256., seed=42)
    b = UniformCatalog(nbar=3e-4, BoxSize=256., seed=43)
    r = FFTPower(a, mode='1d', Nmesh=32, second=b)
    pa = numpy.asarray(a['Position'], dtype='f8')
    pb = numpy.asarray(b['Position'], dtype='f8')
    want = fftpower_oracle(pa, second_position=pb, Nmesh=32, BoxSize=256.,
                           mode='1d', resampler='cic', compensated=True)
    check_parity(r, want)
    assert r.attrs['shotnoise'] == 0.0


def test_lognormal_parity():
    Plin = LinearPower(Planck15, redshift=0.55)
    cat = LogNormalCatalog(Plin=Plin, nbar=2e-4, BoxSize=256., Nmesh=64,
                           bias=2.0, seed=42)
    r = FFTPower(cat, mode='1d', Nmesh=64)
    pos = numpy.asarray(cat['Position'], dtype='f8')
    want = fftpower_oracle(pos, Nmesh=64, BoxSize=256., mode='1d',
                           resampler='cic', compensated=True)
    check_parity(r, want)


def test_dk0_unique_edges_parity():
    cat = UniformCatalog(nbar=3e-4, BoxSize=512., seed=42)
    r = FFTPower(cat, mode='1d', Nmesh=32, dk=0)
    assert_allclose(r.power.coords['k'], r.power['k'], rtol=1e-6)


def test_zero_mode_cleared():
    cat = UniformCatalog(nbar=3e-4, BoxSize=512., seed=42)
    r = FFTPower(cat, mode='1d', Nmesh=32)
    assert_array_equal(r.power['power'][0], 0)


def test_chunked_paint_invariance():
    cat = UniformCatalog(nbar=3e-4, BoxSize=512., seed=42)
    mesh = cat.to_mesh(Nmesh=64, resampler='tsc', interlaced=True,
                       compensated=True, dtype='f8')
    with set_options(paint_chunk_size=cat.csize // 4):
        r1 = mesh.compute()
    with set_options(paint_chunk_size=cat.csize):
        r2 = mesh.compute()
    assert_allclose(r1[...], r2[...], rtol=1e-11, atol=1e-11)
    assert_allclose(r1.attrs['shotnoise'], r2.attrs['shotnoise'])


def test_paint_empty_gpu():
    cat = UniformCatalog(nbar=3e-4, BoxSize=512., seed=42)
    sub = cat[:0]
    assert sub.csize == 0
    mesh = sub.to_mesh(Nmesh=32, resampler='tsc', interlaced=True,
                       compensated=True, position='Position')
    real = mesh.to_real_field(normalize=True)
    assert_allclose(real[...], 1.0)
    real = mesh.to_real_field(normalize=False)
    assert_allclose(real[...], 0.0)


def test_save_load_roundtrip(tmp_path):
    cat = UniformCatalog(nbar=3e-4, BoxSize=512., seed=42)
    r = FFTPower(cat, mode='2d', Nmesh=32)
    path = str(tmp_path / 'fftpower-test.json')
    r.save(path)
    r2 = FFTPower.load(path)
    assert_array_equal(r.power['k'], r2.power['k'])
    assert_array_equal(r.power['power'], r2.power['power'])
    assert_array_equal(r.power['mu'], r2.power['mu'])
    assert_array_equal(r.power['modes'], r2.power['modes'])


def test_fftpower_padding():
    source = UniformCatalog(nbar=3e-4, BoxSize=512., seed=42)
    r = FFTPower(source, mode='1d', BoxSize=1024, Nmesh=32)
    assert r.attrs['N1'] != 0 and r.attrs['N2'] != 0


# ---- full-size property tests on the GPU (the reference's own suite,
# at the reference's sizes: test_fftpower.py:12-44 etc.) -----------------

@pytest.mark.parametrize('window', ['cic', 'tsc'])
def test_aliasing_property_gpu(window):
    source = UniformCatalog(nbar=3e-4, BoxSize=512., seed=42)
    mesh = source.to_mesh(resampler=window, Nmesh=64, compensated=True)
    r = FFTPower(mesh, mode='1d', kmin=0.02)
    Pk = r.power['power'].real
    sel = ~numpy.isnan(Pk)
    err = (2 * Pk[sel] ** 2 / r.power['modes'][sel]) ** 0.5
    red_chi2 = (((Pk[sel] - r.attrs['shotnoise']) / err) ** 2).sum() \
        / sel.sum()
    assert red_chi2 < 1.0


def test_poles_identity_gpu():
    source = UniformCatalog(nbar=3e-3, BoxSize=512., seed=42)
    r = FFTPower(source, mode='2d', BoxSize=1024, Nmesh=32,
                 poles=[0, 2, 4])
    pkmu = r.power['power'].real
    mono = r.poles['power_0'].real
    modes_1d = r.power['modes'].sum(axis=-1)
    mono_from_pkmu = numpy.nansum(pkmu * r.power['modes'], axis=-1) \
        / modes_1d
    assert_array_equal(modes_1d, r.poles['modes'])
    assert_allclose(mono_from_pkmu, mono)


def test_device_catalog_resident_path():
    """GPU-resident positions (the bench input path) give the same
    result as host columns."""
    from nbodykit_amd.source.catalog.device import DeviceArrayCatalog
    cat = UniformCatalog(nbar=3e-4, BoxSize=256., seed=42)
    pos = numpy.asarray(cat['Position'], dtype='f8')
    dcat = DeviceArrayCatalog(
        {'Position': torch.as_tensor(pos).to('cuda')}, BoxSize=cat.attrs['BoxSize'])
    r1 = FFTPower(cat, mode='1d', Nmesh=32)
    r2 = FFTPower(dcat, mode='1d', Nmesh=32)
    assert_array_equal(r1.power['modes'], r2.power['modes'])
    assert_allclose(numpy.nan_to_num(r1.power['power']),
                    numpy.nan_to_num(r2.power['power']),
                    rtol=1e-12, atol=1e-12)


# ---- FFTCorr (xi(r), the first SURVEY §8f widening row) ----------------

def test_fftcorr_parity():
    from nbodykit_amd.lab import FFTCorr
    from oracle import fftcorr_oracle
    cat = UniformCatalog(nbar=3e-4, BoxSize=256., seed=42)
    r = FFTCorr(cat, mode='1d', Nmesh=32)
    pos = uniform_positions(3e-4, 256., 42)
    want = fftcorr_oracle(pos, Nmesh=32, BoxSize=256., mode='1d')
    assert_array_equal(r.corr['modes'], want['modes'])
    got = r.corr['corr']
    ref = want['corr']
    ok = numpy.isfinite(ref) & (numpy.abs(ref) > 1e-12)
    rel = numpy.abs(got[ok] - ref[ok]) / numpy.abs(ref[ok])
    assert rel.max() < PARITY_RTOL, 'xi parity: %g' % rel.max()
    assert_allclose(numpy.nan_to_num(r.corr['r']),
                    numpy.nan_to_num(want['r']), rtol=1e-10, atol=1e-12)


def test_fftcorr_parity_2d_poles():
    from nbodykit_amd.lab import FFTCorr
    from oracle import fftcorr_oracle
    cat = UniformCatalog(nbar=1e-3, BoxSize=256., seed=7)
    r = FFTCorr(cat, mode='2d', Nmesh=32, Nmu=4, poles=[0, 2])
    pos = numpy.asarray(cat['Position'], dtype='f8')
    want = fftcorr_oracle(pos, Nmesh=32, BoxSize=256., mode='2d', Nmu=4,
                          poles=[0, 2])
    assert_array_equal(r.corr['modes'], want['modes'])
    got = numpy.ravel(r.corr['corr'])
    ref = numpy.ravel(want['corr'])
    ok = numpy.isfinite(ref) & (numpy.abs(ref) > 1e-12)
    rel = numpy.abs(got[ok] - ref[ok]) / numpy.abs(ref[ok])
    assert rel.max() < PARITY_RTOL
    for ell in (0, 2):
        got = r.poles['corr_%d' % ell]
        ref = want['poles'][ell]
        ok = numpy.isfinite(ref) & (numpy.abs(ref) > 1e-12)
        rel = numpy.abs(got[ok] - ref[ok]) / numpy.abs(ref[ok])
        assert rel.max() < PARITY_RTOL, 'xi_%d parity: %g' % (ell, rel.max())


def test_fftcorr_save_load(tmp_path):
    from nbodykit_amd.lab import FFTCorr
    cat = UniformCatalog(nbar=3e-4, BoxSize=256., seed=42)
    r = FFTCorr(cat, mode='1d', Nmesh=32)
    path = str(tmp_path / 'fftcorr.json')
    r.save(path)
    r2 = FFTCorr.load(path)
    assert_array_equal(r.corr['corr'], r2.corr['corr'])


# ---- readout + FFTRecon ------------------------------

def test_readout_kernel_parity():
    from nbodykit_amd.pm import ParticleMesh, RealField
    from oracle import MeshGeometry, readout as oracle_readout
    geom = MeshGeometry(16, 32.)
    rng = numpy.random.RandomState(21)
    mesh = rng.normal(size=(16, 16, 16))
    pos = rng.uniform(0, 32., size=(3000, 3))

    pm = ParticleMesh(BoxSize=32., Nmesh=16)
    field = RealField(pm, tensor=torch.as_tensor(mesh).to('cuda'))
    pos_t = torch.as_tensor(pos).to('cuda')
    for resampler in ('cic', 'tsc', 'pcs', 'nnb'):
        got = field.readout(pos_t, resampler=resampler).cpu().numpy()
        want = oracle_readout(pos, mesh, geom, resampler=resampler)
        assert_allclose(got, want, rtol=1e-12, atol=1e-12)


def test_fftrecon_parity():
    from nbodykit_amd.lab import FFTRecon
    from oracle import fftrecon_oracle
    data = UniformCatalog(nbar=2e-3, BoxSize=64., seed=9)
    ran = UniformCatalog(nbar=4e-3, BoxSize=64., seed=10)
    dpos = numpy.asarray(data['Position'], dtype='f8')
    rpos = numpy.asarray(ran['Position'], dtype='f8')
    for scheme in ('LGS', 'LF2', 'LRR'):
        recon = FFTRecon(data, ran, Nmesh=16, BoxSize=64., bias=1.2,
                         f=0.3, R=10.0, scheme=scheme)
        got = numpy.asarray(recon.compute(mode='real'))
        want = fftrecon_oracle(dpos, rpos, Nmesh=16, BoxSize=64.,
                               bias=1.2, f=0.3, R=10.0, scheme=scheme)
        assert_allclose(got, want, rtol=1e-9, atol=1e-11)


def test_fftrecon_through_fftpower():
    from nbodykit_amd.lab import FFTRecon
    data = UniformCatalog(nbar=2e-3, BoxSize=128., seed=9)
    ran = UniformCatalog(nbar=4e-3, BoxSize=128., seed=10)
    recon = FFTRecon(data, ran, Nmesh=32, BoxSize=128., R=16.0)
    r = FFTPower(recon, mode='1d')
    assert numpy.isfinite(r.power['power'].real[1:]).any()


def test_projected_fftpower():
    """ProjectedFFTPower (fftpower.py:361-505): projecting over all
    axes of a 3D field reduces to rfftn of the summed preview; check
    against a direct numpy restatement on the same painted field."""
    from nbodykit_amd.lab import ProjectedFFTPower
    cat = UniformCatalog(nbar=1e-3, BoxSize=128., seed=5)
    mesh = cat.to_mesh(Nmesh=32, dtype='f8')
    r = ProjectedFFTPower(mesh, axes=(0, 1))

    field = mesh.compute(mode='real')
    r1 = numpy.asarray(field).sum(axis=2)
    c1 = numpy.fft.rfftn(r1) / 32 ** 3
    pk = (c1 * c1.conj())
    pk.flat[0] = 0
    # same binning math as the class
    shape = numpy.array([32, 32])
    box = numpy.array([128., 128.])
    I = numpy.eye(2, dtype='int') * -2 + 1
    k = [numpy.fft.fftfreq(N, 1. / (N * 2 * numpy.pi / L))[:pkshape]
         .reshape(kshape)
         for N, L, kshape, pkshape in zip(shape, box, I, pk.shape)]
    kmag = sum(ki ** 2 for ki in k) ** 0.5
    W = numpy.empty(pk.shape, dtype='f4')
    W[...] = 2.0
    W[..., 0] = 1.0
    W[..., -1] = 1.0
    dk = 2 * numpy.pi / 128.
    kedges = numpy.arange(0., numpy.pi * 32 / 128. + dk / 2, dk)
    Nsum = numpy.zeros(len(kedges) + 1)
    Psum = numpy.zeros(len(kedges) + 1, dtype='c16')
    dig = numpy.digitize(kmag.flat, kedges)
    Psum.real.flat += numpy.bincount(dig, weights=(W * pk.real).flat,
                                     minlength=Nsum.size)
    Nsum.flat += numpy.bincount(dig, weights=W.flat, minlength=Nsum.size)
    with numpy.errstate(invalid='ignore', divide='ignore'):
        want = (Psum / Nsum)[1:-1] * box.prod()
    got = r.power['power']
    ok = numpy.isfinite(want.real) & (numpy.abs(want) > 0)
    assert_allclose(got[ok].real, want[ok].real, rtol=1e-10)


def test_lab_end_to_end(tmp_path):
    """The reference's full-pipeline smoke (nbodykit/tests/test_lab.py:
    11-27): EH LinearPower -> LogNormalCatalog -> RSD shift via
    transform.VectorProjection -> FFTPower(mode='2d', poles=[0,2,4]) ->
    JSON save/load."""
    from nbodykit_amd.lab import cosmology, transform
    from nbodykit_amd.lab import LogNormalCatalog as LNC
    cosmo = cosmology.Planck15
    Plin = cosmology.LinearPower(cosmo, redshift=0.55,
                                 transfer='EisensteinHu')
    source = LNC(Plin=Plin, nbar=3e-5, BoxSize=690., Nmesh=16, seed=42)

    source['Position'] = numpy.asarray(source['Position']) \
        + transform.VectorProjection(source['VelocityOffset'], [0, 0, 1])

    result = FFTPower(source, mode='2d', Nmesh=32, poles=[0, 2, 4],
                      los=[0, 0, 1])
    out = str(tmp_path / 'test_fftpower.json')
    result.save(out)
    back = FFTPower.load(out)
    assert_array_equal(result.poles['power_2'], back.poles['power_2'])
    assert result.attrs['N1'] == source.csize


def test_compute_resample():
    """compute(Nmesh=...) resamples by Fourier-mode copy (base/
    mesh.py:320-330): large-scale modes preserved, mean exact."""
    from oracle import MeshGeometry, r2c as oracle_r2c
    cat = UniformCatalog(nbar=1e-3, BoxSize=128., seed=5)
    mesh = cat.to_mesh(Nmesh=64, dtype='f8')
    full = mesh.compute(mode='real')
    down = mesh.compute(mode='real', Nmesh=32)
    assert numpy.asarray(down).shape == (32, 32, 32)
    assert_allclose(down.cmean(), full.cmean(), rtol=1e-10)

    # modes below the new Nyquist agree with the oracle truncation
    geom64 = MeshGeometry(64, 128.)
    cfull = oracle_r2c(numpy.asarray(full), geom64)
    cdown = oracle_r2c(numpy.asarray(down), MeshGeometry(32, 128.))
    assert_allclose(cdown[:8, :8, :8], cfull[:8, :8, :8],
                    rtol=1e-10, atol=1e-13)

    up = mesh.compute(mode='complex', Nmesh=128)
    assert up.cshape == (128, 128, 65)


# ---------------------------------------------------------------------------
# ConvolvedFFTPower (survey-geometry FKP multipoles)
# ---------------------------------------------------------------------------

def _survey_mock(seed=99, ndata=3000, nran=30000):
    """A survey-like mock: data+randoms filling a sub-box far from the
    origin (so xhat varies across the volume), constant n(z)."""
    from nbodykit_amd.lab import ArrayCatalog
    rng = numpy.random.RandomState(seed)
    lo = numpy.array([1000., 1200., 900.])
    span = numpy.array([300., 260., 340.])
    nbar = ndata / span.prod()
    data = ArrayCatalog({
        'Position': lo + rng.uniform(0., 1., size=(ndata, 3)) * span,
        'NZ': numpy.full(ndata, nbar)})
    ran = ArrayCatalog({
        'Position': lo + rng.uniform(0., 1., size=(nran, 3)) * span,
        'NZ': numpy.full(nran, nbar)})
    return data, ran


def test_convpower_parity():
    """Product ConvolvedFFTPower vs the oracle restatement (which uses
    an independent scipy formulation of the Ylm)."""
    from nbodykit_amd.lab import FKPCatalog, ConvolvedFFTPower
    from oracle.convpower import convpower_oracle

    data, ran = _survey_mock()
    cat = FKPCatalog(data, ran, P0=1e4, BoxSize=400., BoxPad=0.02)
    with pytest.warns(UserWarning):
        mesh = cat.to_mesh(Nmesh=64, BoxCenter=[1150., 1330., 1070.],
                           dtype='c16', compensated=True)
    r = ConvolvedFFTPower(mesh, poles=[0, 2, 4], dk=0.05)

    fkp_d = 1.0 / (1.0 + 1e4 * numpy.asarray(data['NZ']))
    fkp_r = 1.0 / (1.0 + 1e4 * numpy.asarray(ran['NZ']))
    o = convpower_oracle(numpy.asarray(data['Position']),
                         numpy.asarray(ran['Position']),
                         [0, 2, 4], Nmesh=64, BoxSize=400.,
                         BoxCenter=[1150., 1330., 1070.],
                         nbar_data=numpy.asarray(data['NZ']),
                         nbar_ran=numpy.asarray(ran['NZ']),
                         data_fkp=fkp_d, ran_fkp=fkp_r,
                         compensated=True, dk=0.05)

    assert_allclose(r.attrs['alpha'], o['attrs']['alpha'], rtol=1e-12)
    assert_allclose(r.attrs['shotnoise'], o['attrs']['shotnoise'],
                    rtol=1e-10)
    assert_allclose(r.attrs['randoms.norm'], o['attrs']['randoms.norm'],
                    rtol=1e-10)
    assert_array_equal(r.poles['modes'], o['modes'])
    assert_allclose(r.poles['k'], o['k'], rtol=1e-10, equal_nan=True)
    # power_ell stored as c8 in both (the reference's dtype); compare
    # against the scale of the monopole
    scale = numpy.nanmax(numpy.abs(o['power_0']))
    for ell in [0, 2, 4]:
        assert_allclose(r.poles['power_%d' % ell], o['power_%d' % ell],
                        atol=2e-5 * scale, rtol=2e-5, equal_nan=True,
                        err_msg='ell=%d' % ell)


def test_convpower_monopole_shotnoise_gpu():
    """Unclustered sample: P0 - Pshot should scatter around zero."""
    from nbodykit_amd.lab import FKPCatalog, ConvolvedFFTPower
    data, ran = _survey_mock(seed=3)
    cat = FKPCatalog(data, ran, BoxSize=400.)
    with pytest.warns(UserWarning):
        mesh = cat.to_mesh(Nmesh=64, BoxCenter=[1150., 1330., 1070.],
                           compensated=True)
    r = ConvolvedFFTPower(mesh, poles=[0], dk=0.05)
    P0 = r.poles['power_0'].real - r.attrs['shotnoise']
    assert abs(numpy.nanmean(P0)) < 0.5 * r.attrs['shotnoise']


def test_convpower_save_load(tmp_path):
    from nbodykit_amd.lab import FKPCatalog, ConvolvedFFTPower
    data, ran = _survey_mock(seed=17, ndata=500, nran=5000)
    cat = FKPCatalog(data, ran, BoxSize=400.)
    with pytest.warns(UserWarning):
        mesh = cat.to_mesh(Nmesh=32, BoxCenter=[1150., 1330., 1070.])
    r = ConvolvedFFTPower(mesh, poles=[0, 2], dk=0.05)
    fn = str(tmp_path / 'conv.json')
    r.save(fn)
    r2 = ConvolvedFFTPower.load(fn)
    assert_allclose(r2.poles['power_0'], r.poles['power_0'],
                    equal_nan=True)
    assert_allclose(r2.attrs['shotnoise'], r.attrs['shotnoise'])
    # to_pkmu inversion runs
    pkmu = r.to_pkmu(numpy.linspace(0, 1, 3), 2)
    assert pkmu['power'].shape[1] == 2


# ---------------------------------------------------------------------------
# bigfile mesh save/load (reference base/mesh.py:444-500,
# source/mesh/bigfile.py:16-137)
# ---------------------------------------------------------------------------

def test_mesh_save_load_real(tmp_path):
    from nbodykit_amd.lab import UniformCatalog, BigFileMesh, FFTPower
    path = str(tmp_path / 'mesh_r')
    cat = UniformCatalog(nbar=2e-3, BoxSize=64., seed=21)
    mesh = cat.to_mesh(Nmesh=32, dtype='f8', compensated=False)
    rfield = mesh.compute(mode='real')
    mesh.save(path, dataset='Field', mode='real')

    loaded = BigFileMesh(path, 'Field')
    assert not loaded.isfourier
    rfield2 = loaded.compute(mode='real')
    assert_allclose(rfield2.value.cpu().numpy(),
                    rfield.value.cpu().numpy(), rtol=1e-15)
    # attrs survive (shotnoise etc. + the pm vectors)
    assert_allclose(loaded.attrs['shotnoise'], rfield.attrs['shotnoise'])
    assert_array_equal(loaded.attrs['Nmesh'], 32)

    # and the loaded mesh feeds FFTPower like the original
    r1 = FFTPower(mesh, mode='1d')
    r2 = FFTPower(loaded, mode='1d')
    assert_allclose(r2.power['power'], r1.power['power'],
                    rtol=1e-12, equal_nan=True)


def test_mesh_save_load_complex(tmp_path):
    from nbodykit_amd.lab import UniformCatalog, BigFileMesh
    path = str(tmp_path / 'mesh_c')
    cat = UniformCatalog(nbar=2e-3, BoxSize=64., seed=22)
    mesh = cat.to_mesh(Nmesh=32, dtype='f8')
    cfield = mesh.compute(mode='complex')
    mesh.save(path, dataset='Field', mode='complex')

    loaded = BigFileMesh(path, 'Field')
    assert loaded.isfourier
    cfield2 = loaded.compute(mode='complex')
    assert_allclose(cfield2.value.cpu().numpy(),
                    cfield.value.cpu().numpy(), rtol=1e-15)
    # complex -> real via the action pipeline's automatic c2r
    r2 = loaded.compute(mode='real')
    r1 = mesh.compute(mode='real')
    assert_allclose(r2.value.cpu().numpy(), r1.value.cpu().numpy(),
                    rtol=1e-12, atol=1e-12)


def test_catalog_bigfile_fftpower_roundtrip(tmp_path):
    # save a catalog, reload it, and verify FFTPower is identical
    from nbodykit_amd.lab import (UniformCatalog, BigFileCatalog,
                                  FFTPower)
    path = str(tmp_path / 'cat')
    cat = UniformCatalog(nbar=1e-2, BoxSize=32., seed=33)
    cat.save(path)
    loaded = BigFileCatalog(path)
    r1 = FFTPower(cat, mode='1d', Nmesh=32)
    r2 = FFTPower(loaded, mode='1d', Nmesh=32)
    assert_allclose(r2.power['power'], r1.power['power'],
                    rtol=1e-13, equal_nan=True)
    assert_array_equal(r2.power['modes'], r1.power['modes'])


# ---------------------------------------------------------------------------
# fused compensate+power+bin kernel (nbk_power_bin_f64)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize('kwargs', [
    dict(mode='1d'),
    dict(mode='2d', Nmu=5, poles=[0, 2, 4]),
])
def test_fused_power_matches_unfused(kwargs):
    """FFTPower takes the fused single-pass path for plain CatalogMesh
    inputs; a no-op apply() view forces the explicit
    compensate/power3d/bin sequence — results must agree to roundoff."""
    from nbodykit_amd.lab import UniformCatalog, FFTPower
    cat = UniformCatalog(nbar=1e-2, BoxSize=64., seed=7)
    mesh = cat.to_mesh(Nmesh=64, dtype='f8', compensated=True,
                       resampler='tsc', interlaced=True)
    r_fused = FFTPower(mesh, **kwargs)
    noop = mesh.apply(lambda x, v: v, kind='wavenumber', mode='complex')
    r_plain = FFTPower(noop, **kwargs)
    # tolerances allow the atomic/run-merged accumulation ordering
    # roundoff and the fast kernel's per-axis-product compensation
    # composition (vs chained divides) — last-ulp differences
    scale = numpy.nanmax(numpy.abs(r_plain.power['power']))
    assert_allclose(r_fused.power['power'], r_plain.power['power'],
                    rtol=1e-10, atol=1e-12 * scale, equal_nan=True)
    assert_array_equal(r_fused.power['modes'], r_plain.power['modes'])
    if 'poles' in kwargs:
        for ell in kwargs['poles']:
            assert_allclose(r_fused.poles['power_%d' % ell],
                            r_plain.poles['power_%d' % ell],
                            rtol=1e-10, atol=1e-12 * scale,
                            equal_nan=True)


def test_fused_cross_power_matches_unfused():
    from nbodykit_amd.lab import UniformCatalog, FFTPower
    cat1 = UniformCatalog(nbar=1e-2, BoxSize=64., seed=8)
    cat2 = UniformCatalog(nbar=1e-2, BoxSize=64., seed=9)
    m1 = cat1.to_mesh(Nmesh=32, dtype='f8', compensated=True)
    m2 = cat2.to_mesh(Nmesh=32, dtype='f8', compensated=True,
                      resampler='pcs')
    r_fused = FFTPower(m1, mode='1d', second=m2)
    r_plain = FFTPower(m1.apply(lambda x, v: v, mode='complex'),
                       mode='1d',
                       second=m2.apply(lambda x, v: v, mode='complex'))
    scale = numpy.nanmax(numpy.abs(r_plain.power['power']))
    assert_allclose(r_fused.power['power'], r_plain.power['power'],
                    rtol=1e-10, atol=1e-12 * scale, equal_nan=True)


# ---------------------------------------------------------------------------
# ArrayMesh / LinearMesh (reference source/mesh/array.py, linear.py)
# ---------------------------------------------------------------------------

def test_arraymesh_fftpower_parity():
    from nbodykit_amd.lab import ArrayMesh, FFTPower
    from oracle.mesh import MeshGeometry, r2c
    from oracle.fftpower import project_to_basis as oracle_project
    rng = numpy.random.RandomState(31)
    arr = 1.0 + 0.1 * rng.standard_normal((32, 32, 32))
    mesh = ArrayMesh(arr, BoxSize=64.)
    got = mesh.compute(mode='real')
    assert_allclose(got.value.cpu().numpy(), arr, rtol=1e-15)

    r = FFTPower(mesh, mode='1d')
    geom = MeshGeometry([32] * 3, 64.)
    c = r2c(arr, geom)
    p3d = c * numpy.conj(c) * 64. ** 3
    p3d.flat[0] = 0.0
    (xm, _, y2d, N2d), _ = oracle_project(
        p3d, geom, [r.power.edges['k'], numpy.linspace(-1, 1, 2)])
    assert_allclose(r.power['power'], numpy.squeeze(y2d),
                    rtol=1e-10, equal_nan=True)
    assert_array_equal(r.power['modes'], numpy.squeeze(N2d))


def test_linearmesh_unitary_flat_power():
    # unitary_amplitude fixes |delta(k)|^2 = P/V exactly, so the
    # measured P(k) is exactly P0 in every populated bin
    from nbodykit_amd.lab import LinearMesh, FFTPower
    P0 = 123.0
    mesh = LinearMesh(lambda k: P0 * numpy.ones_like(k), BoxSize=128.,
                      Nmesh=32, seed=11, unitary_amplitude=True)
    r = FFTPower(mesh, mode='1d')
    p = r.power['power'].real
    # the first bin contains the cleared k=0 mode (binned as 0 with
    # weight 1 — the reference's documented behavior), lowering its mean
    good = ~numpy.isnan(p)
    good[0] = False
    assert_allclose(p[good], P0, rtol=1e-10)
    nmodes0 = r.power['modes'][0]
    assert_allclose(p[0] * nmodes0, P0 * (nmodes0 - 1), rtol=1e-10)


def test_linearmesh_inverted_phase():
    from nbodykit_amd.lab import LinearMesh
    kw = dict(BoxSize=128., Nmesh=16, seed=5)
    P = lambda k: 10.0 * numpy.ones_like(k)
    m1 = LinearMesh(P, **kw).compute(mode='real')
    m2 = LinearMesh(P, inverted_phase=True, **kw).compute(mode='real')
    total = m1.value + m2.value
    assert_allclose(total.cpu().numpy(), 2.0, rtol=1e-10)


def test_linearmesh_through_fftpower_statistics():
    # non-unitary realization: bin means scatter around P0 with
    # ~1/sqrt(Nmodes) errors; loose statistical bound
    from nbodykit_amd.lab import LinearMesh, FFTPower
    P0 = 50.0
    mesh = LinearMesh(lambda k: P0 * numpy.ones_like(k), BoxSize=256.,
                      Nmesh=64, seed=4)
    r = FFTPower(mesh, mode='1d')
    p = r.power['power'].real
    N = r.power['modes']
    ok = N > 50
    chi = (p[ok] - P0) / (P0 * numpy.sqrt(2.0 / N[ok]))
    assert numpy.abs(chi).max() < 5.0


def test_survey_workflow_end_to_end():
    """SkyToCartesian -> FKPCatalog -> ConvolvedFFTPower: the full
    survey pipeline composes (transform + cosmology + FKP)."""
    from nbodykit_amd.lab import (transform, ArrayCatalog, FKPCatalog,
                                  ConvolvedFFTPower)
    from nbodykit_amd.cosmology import Planck15
    rng = numpy.random.RandomState(77)

    def make(n):
        ra = rng.uniform(20., 40., n)
        dec = rng.uniform(-10., 10., n)
        z = rng.uniform(0.4, 0.6, n)
        pos = transform.SkyToCartesian(ra, dec, z, Planck15)
        return ArrayCatalog({'Position': pos,
                             'NZ': numpy.full(n, 3e-4)})

    data, ran = make(2000), make(20000)
    cat = FKPCatalog(data, ran, P0=1e4)
    with pytest.warns(UserWarning):
        mesh = cat.to_mesh(Nmesh=32, dtype='c16', compensated=True)
    r = ConvolvedFFTPower(mesh, poles=[0, 2], dk=0.05)
    P0m = r.poles['power_0'].real - r.attrs['shotnoise']
    assert numpy.isfinite(r.attrs['shotnoise'])
    assert numpy.nanmax(numpy.abs(P0m)) < 50 * r.attrs['shotnoise']
    # bbox derived from the randoms covers the data
    assert (numpy.asarray(r.attrs['BoxSize']) > 0).all()


def test_fused_complex_paint_matches_real_path():
    """CatalogMesh.to_complex_field (paint with the z-FFT fused into the
    tile flush) must equal r2c of the real path, plain and interlaced,
    once the two-level thresholds admit the gather kernel."""
    import torch
    from nbodykit_amd import set_options
    from nbodykit_amd.lab import ArrayCatalog
    rng = numpy.random.RandomState(41)
    n = 180000
    cat = ArrayCatalog({'Position': rng.uniform(0, 64., size=(n, 3)),
                        'Weight': rng.exponential(size=n)})
    for interlaced, resampler in [(False, 'cic'), (True, 'tsc'),
                                  (False, 'pcs')]:
        kw = dict(Nmesh=64, BoxSize=64., dtype='f8', resampler=resampler,
                  interlaced=interlaced, compensated=False)
        with set_options(sort_min_n=1024, sort_two_level_min_n=1024,
                         sort_two_level_min_cells=1):
            mesh = cat.to_mesh(**kw)
            c_fused = mesh.to_complex_field()
            assert c_fused is not NotImplemented
            c_ref = mesh.to_real_field().r2c()
        a = c_fused.value.cpu().numpy()
        b = c_ref.value.cpu().numpy()
        scale = numpy.abs(b).max()
        assert_allclose(a, b, atol=1e-12 * scale, rtol=1e-10,
                        err_msg='interlaced=%s %s' % (interlaced,
                                                      resampler))
        for key in ['N', 'W', 'shotnoise']:
            assert_allclose(c_fused.attrs[key], c_ref.attrs[key],
                            rtol=1e-12)


def test_fused_complex_paint_through_fftpower():
    # the whole FFTPower pipeline rides the fused path when thresholds
    # allow; results must agree with the unfused pipeline
    from nbodykit_amd import set_options
    from nbodykit_amd.lab import UniformCatalog, FFTPower
    cat = UniformCatalog(nbar=0.5, BoxSize=64., seed=3)   # ~1.3e5 pts
    kw = dict(mode='2d', Nmu=4, poles=[0, 2], Nmesh=64)
    r_plain = FFTPower(cat, **kw)
    with set_options(sort_min_n=1024, sort_two_level_min_n=1024,
                     sort_two_level_min_cells=1):
        r_fused = FFTPower(cat, **kw)
    scale = numpy.nanmax(numpy.abs(r_plain.power['power']))
    assert_allclose(r_fused.power['power'], r_plain.power['power'],
                    rtol=1e-10, atol=1e-11 * scale, equal_nan=True)
    assert_array_equal(r_fused.power['modes'], r_plain.power['modes'])
    assert_allclose(r_fused.attrs['shotnoise'],
                    r_plain.attrs['shotnoise'], rtol=1e-12)


def test_mesh_filters():
    """TopHat/Gaussian MeshFilters through apply() (reference
    filters.py + tests/test_filters.py): the Gaussian-filtered power is
    P(k) exp(-k^2 r^2)."""
    from nbodykit_amd.lab import LinearMesh, FFTPower, filters
    P0 = 100.0
    mesh = LinearMesh(lambda k: P0 * numpy.ones_like(k), BoxSize=256.,
                      Nmesh=64, seed=9, unitary_amplitude=True)
    sm = mesh.apply(filters.Gaussian(8.0))
    r = FFTPower(sm, mode='1d')
    k = r.power['k']
    p = r.power['power'].real
    # restrict to kr < 2: beyond that the bin-average of the damped
    # exponential differs from its value at the mean k (Jensen)
    good = numpy.isfinite(p) & (k > 0) & (r.power['modes'] > 8) \
        & (k * 8.0 < 2.0)
    expect = P0 * numpy.exp(-(k[good] * 8.0) ** 2)
    assert good.sum() >= 5
    assert_allclose(p[good], expect, rtol=0.1)
    # TopHat runs and leaves the k=0 normalization intact
    th = mesh.apply(filters.TopHat(8.0)).compute(mode='real')
    assert abs(th.value.mean().item() - 1.0) < 1e-10


def test_linearmesh_chisq_vs_theory():
    """reference source/mesh/tests/test_linear.py:12-35: P(k) measured
    from a LinearMesh realization agrees with the input Plin at reduced
    chi^2 < 1.5 (variance 2 P^2 / Nmodes per bin)."""
    from nbodykit_amd.lab import LinearMesh, FFTPower, LinearPower
    from nbodykit_amd.cosmology import Planck15
    Plin = LinearPower(Planck15, redshift=0.55,
                       transfer='EisensteinHu')
    source = LinearMesh(Plin, Nmesh=64, BoxSize=512., seed=42)
    r = FFTPower(source, mode='1d', Nmesh=64, dk=0.01, kmin=0.005)
    valid = r.power['modes'] > 0
    theory = Plin(r.power['k'][valid])
    errs = (2 * theory ** 2 / r.power['modes'][valid]) ** 0.5
    chisq = (((r.power['power'][valid].real - theory) / errs) ** 2)
    red = chisq.sum() / (valid.sum() - 1)
    assert red < 1.5, red


def test_species_mesh_summed_paint():
    """MultipleSpeciesCatalogMesh sums species densities with combined
    shot noise (reference source/mesh/tests/test_species.py:50-96)."""
    from nbodykit_amd.lab import (ArrayCatalog, MultipleSpeciesCatalog,
                                  FFTPower)
    rng = numpy.random.RandomState(21)
    c1 = ArrayCatalog({'Position': rng.uniform(0, 64., size=(4000, 3))})
    c2 = ArrayCatalog({'Position': rng.uniform(0, 64., size=(8000, 3))})
    cat = MultipleSpeciesCatalog(['a', 'b'], c1, c2)
    mesh = cat.to_mesh(Nmesh=32, BoxSize=64., dtype='f8')
    real = mesh.compute(mode='real')

    # the summed raw density equals painting the concatenated catalog
    both = ArrayCatalog({'Position': numpy.concatenate(
        [numpy.asarray(c1['Position']), numpy.asarray(c2['Position'])])})
    want = both.to_mesh(Nmesh=32, BoxSize=64., dtype='f8') \
        .compute(mode='real')
    assert_allclose(real.value.cpu().numpy(), want.value.cpu().numpy(),
                    rtol=1e-12, atol=1e-12)

    # attrs: N summed, per-species prefixes, combined shotnoise
    assert real.attrs['N'] == 12000
    assert real.attrs['a.N'] == 4000 and real.attrs['b.N'] == 8000
    expect_sn = sum((real.attrs['%s.W' % s] / 12000.) ** 2
                    * real.attrs['%s.shotnoise' % s] for s in 'ab')
    assert_allclose(real.attrs['shotnoise'], expect_sn, rtol=1e-12)

    # and it flows through FFTPower
    r = FFTPower(mesh, mode='1d')
    assert numpy.isfinite(r.power['power'].real[1:]).any()


# ---- at-scale parity: production-size inputs
# through the production thresholds, against the oracle at 1e-10 --------

@pytest.mark.timeout(900)
def test_at_scale_c2_parity():
    """C2 shape (1e7 uniform pts / 256^3, CIC compensated) at DEFAULT
    gates: the two-level sort + ownership-gather + fused z-FFT path as
    the bench runs it.  The small-mesh tests can't see scale-dependent
    index/digitize/sort-threshold bugs (oracle ~11 s CPU)."""
    n = int(1e7)
    pos = numpy.random.RandomState(42).uniform(0, 1000., size=(n, 3))
    cat = ArrayCatalog({'Position': pos})
    mesh = cat.to_mesh(Nmesh=256, BoxSize=1000., dtype='f8',
                       compensated=True, resampler='cic')
    # must take the fused paint+z-FFT path at production thresholds
    assert mesh.to_complex_field() is not NotImplemented
    r = FFTPower(mesh, mode='1d')
    want = fftpower_oracle(pos, Nmesh=256, BoxSize=1000., mode='1d',
                           resampler='cic', compensated=True)
    assert_array_equal(r.power['modes'], want['modes'])
    got = r.power['power'].real
    ref = want['power'].real
    ok = numpy.isfinite(ref) & (numpy.abs(ref) > 0)
    rel = numpy.abs(got[ok] - ref[ok]) / numpy.abs(ref[ok])
    assert rel.max() < 1e-10, 'at-scale P parity: %g' % rel.max()


@pytest.mark.timeout(900)
def test_at_scale_512_tsc_interlaced_spot():
    """512^3 TSC+interlaced spot check (C3's mesh size): big-mesh index
    math, interlaced ghost range and the k-space combine at a
    production mesh (thresholds lowered so 1e6 pts engage the fused
    path; the kernels see the same 512^3 geometry the bench does)."""
    n = int(1e6)
    pos = numpy.random.RandomState(7).uniform(0, 2500., size=(n, 3))
    cat = ArrayCatalog({'Position': pos})
    with set_options(sort_min_n=100000, sort_two_level_min_n=100000,
                     sort_two_level_min_cells=1 << 23):
        mesh = cat.to_mesh(Nmesh=512, BoxSize=2500., dtype='f8',
                           compensated=True, resampler='tsc',
                           interlaced=True)
        assert mesh.to_complex_field() is not NotImplemented
        r = FFTPower(mesh, mode='1d')
    want = fftpower_oracle(pos, Nmesh=512, BoxSize=2500., mode='1d',
                           resampler='tsc', compensated=True,
                           interlaced=True)
    assert_array_equal(r.power['modes'], want['modes'])
    got = r.power['power'].real
    ref = want['power'].real
    ok = numpy.isfinite(ref) & (numpy.abs(ref) > 0)
    rel = numpy.abs(got[ok] - ref[ok]) / numpy.abs(ref[ok])
    assert rel.max() < 1e-10, '512^3 interlaced parity: %g' % rel.max()


def test_nonpow2_mesh_parity():
    """Nmesh=96 (2^5 x 3): the Bluestein fallback composed from the
    power-of-two kernels (pm.fft_axis1 / fft_r2c_z) against the oracle.
    FFTW-backed pmesh accepts arbitrary Nmesh (base/mesh.py:50); every
    BASELINE config is a power of two, so this is capability coverage."""
    n = 50000
    pos = numpy.random.RandomState(3).uniform(0, 300., size=(n, 3))
    cat = ArrayCatalog({'Position': pos})
    mesh = cat.to_mesh(Nmesh=96, BoxSize=300., dtype='f8',
                       compensated=True, resampler='tsc')
    r = FFTPower(mesh, mode='1d')
    want = fftpower_oracle(pos, Nmesh=96, BoxSize=300., mode='1d',
                           resampler='tsc', compensated=True)
    assert_array_equal(r.power['modes'], want['modes'])
    got = r.power['power'].real
    ref = want['power'].real
    ok = numpy.isfinite(ref) & (numpy.abs(ref) > 0)
    rel = numpy.abs(got[ok] - ref[ok]) / numpy.abs(ref[ok])
    assert rel.max() < 1e-9, 'non-pow2 parity: %g' % rel.max()


def test_nonpow2_roundtrip_and_rfftn():
    """96^3 r2c against numpy rfftn and the c2r round trip."""
    import torch
    from nbodykit_amd.pm import ParticleMesh, RealField
    pm = ParticleMesh(BoxSize=100., Nmesh=96)
    rng = numpy.random.RandomState(11)
    arr = rng.normal(size=(96, 96, 96))
    f = RealField(pm, tensor=torch.as_tensor(arr).to('cuda'))
    c = f.r2c()
    want = numpy.fft.rfftn(arr) / 96.0 ** 3
    assert_allclose(c.value.cpu().numpy(), want, rtol=0, atol=1e-13)
    back = c.c2r()
    assert_allclose(back.value.cpu().numpy(), arr, rtol=0, atol=1e-11)


@pytest.mark.parametrize('nmesh', [27, 45])
def test_odd_mesh_parity(nmesh):
    """ODD Nmesh (no Nyquist plane on any axis): full FFTPower vs the
    oracle.  FFTW-backed pmesh accepts any Nmesh; odd lengths run the
    Bluestein fallback with the parity-dependent conventions switched
    (no Nyquist-as-negative, no self-conjugate z plane beyond DC)."""
    cat = UniformCatalog(nbar=2e-3, BoxSize=64., seed=5)
    r = FFTPower(cat, mode='1d', Nmesh=nmesh)
    pos = uniform_positions(2e-3, 64., 5)
    want = fftpower_oracle(pos, Nmesh=nmesh, BoxSize=64., mode='1d',
                           resampler='cic', compensated=True)
    check_parity(r, want)


def test_odd_mesh_interlaced_poles():
    """Odd Nmesh with TSC + interlacing (the k-space combine's
    self-conjugate projection reduces to the DC plane only) and
    multipoles, 2d."""
    n = 30000
    pos = numpy.random.RandomState(7).uniform(0, 120., size=(n, 3))
    cat = ArrayCatalog({'Position': pos})
    mesh = cat.to_mesh(Nmesh=45, BoxSize=120., dtype='f8',
                       compensated=True, resampler='tsc',
                       interlaced=True)
    r = FFTPower(mesh, mode='2d', Nmu=4, poles=[0, 2])
    want = fftpower_oracle(pos, Nmesh=45, BoxSize=120., mode='2d',
                           Nmu=4, poles=[0, 2], resampler='tsc',
                           compensated=True, interlaced=True)
    check_parity(r, want, poles=[0, 2])


def test_odd_roundtrip_and_rfftn():
    """27^3 and 45^3 r2c against numpy rfftn and the c2r round trip."""
    import torch
    from nbodykit_amd.pm import ParticleMesh, RealField
    for nm in (27, 45):
        pm = ParticleMesh(BoxSize=100., Nmesh=nm)
        rng = numpy.random.RandomState(nm)
        arr = rng.normal(size=(nm, nm, nm))
        f = RealField(pm, tensor=torch.as_tensor(arr).to('cuda'))
        c = f.r2c()
        want = numpy.fft.rfftn(arr) / float(nm) ** 3
        assert_allclose(c.value.cpu().numpy(), want, rtol=0, atol=1e-13)
        back = c.c2r()
        assert_allclose(back.value.cpu().numpy(), arr, rtol=0,
                        atol=1e-11)


def test_odd_mesh_fftcorr():
    """Odd-mesh FFTCorr (configuration-space binning + c2r on odd
    lengths) vs the oracle."""
    from nbodykit_amd.lab import FFTCorr
    from oracle import fftcorr_oracle
    n = 20000
    pos = numpy.random.RandomState(9).uniform(0, 100., size=(n, 3))
    cat = ArrayCatalog({'Position': pos})
    r = FFTCorr(cat, mode='1d', Nmesh=33, BoxSize=100.)
    want = fftcorr_oracle(pos, Nmesh=33, BoxSize=100., mode='1d',
                          resampler='cic', compensated=True)
    assert_array_equal(r.corr['modes'], want['modes'])
    got = r.corr['corr']
    ref = want['corr']
    ok = numpy.isfinite(ref) & (numpy.abs(ref) > 1e-12)
    rel = numpy.abs(got[ok] - ref[ok]) / numpy.abs(ref[ok])
    assert rel.max() < 1e-9, 'odd FFTCorr parity: %g' % rel.max()


@pytest.mark.timeout(600)
def test_pcs_midscale_parity():
    """PCS (support 4, 64 deposits/particle) at a 256^3 mesh with the
    gather path engaged — the widest window was previously only
    parity-tested on toy meshes."""
    n = int(5e5)
    pos = numpy.random.RandomState(13).uniform(0, 750., size=(n, 3))
    cat = ArrayCatalog({'Position': pos})
    with set_options(sort_min_n=100000, sort_two_level_min_n=100000,
                     sort_two_level_min_cells=1 << 23):
        mesh = cat.to_mesh(Nmesh=256, BoxSize=750., dtype='f8',
                           compensated=True, resampler='pcs')
        assert mesh.to_complex_field() is not NotImplemented
        r = FFTPower(mesh, mode='2d', Nmu=4, poles=[0, 2])
    want = fftpower_oracle(pos, Nmesh=256, BoxSize=750., mode='2d',
                           Nmu=4, poles=[0, 2], resampler='pcs',
                           compensated=True)
    assert_array_equal(r.power['modes'], want['modes'])
    got = numpy.nan_to_num(r.power['power'].real)
    ref = numpy.nan_to_num(want['power'].real)
    ok = numpy.isfinite(ref) & (numpy.abs(ref) > 0)
    rel = numpy.abs(got[ok] - ref[ok]) / numpy.abs(ref[ok])
    assert rel.max() < 1e-9, 'PCS midscale parity: %g' % rel.max()
    for ell in (0, 2):
        g = r.poles['power_%d' % ell].real
        f = want['poles'][ell].real
        ok = numpy.isfinite(f) & (numpy.abs(f) > 0)
        rel = numpy.abs(g[ok] - f[ok]) / numpy.abs(f[ok])
        assert rel.max() < 1e-9, 'PCS pole %d: %g' % (ell, rel.max())


@pytest.mark.timeout(900)
@pytest.mark.parametrize('seed', list(range(8)) + [51, 55, 63])
def test_random_config_fuzz(seed):
    """Randomized config cases vs the oracle (tests/fuzz_sweep.py):
    mesh size incl. non-pow2, box, window, interlacing, mode, poles,
    weights; every 4th seed is an FFTCorr.  Seeds 51/55/63 pin the
    real-field coordinate-rounding bug the wider sweep caught (the
    configuration lattice sits exactly on r-bin edges, so the kernel
    must reproduce fl(fl(f*L)/N) bit-for-bit)."""
    from tests.fuzz_sweep import run_case
    rel, modes_ok, cfg = run_case(seed)
    assert modes_ok, cfg
    assert rel < 1e-9, (rel, cfg)


@pytest.mark.timeout(900)
def test_2048_mesh_capability():
    """2048^3 capability: beyond the two-level sort's LDS budget
    (DESIGN.md — such meshes take the scatter/single-level paths), the
    pipeline must still be CORRECT.  No oracle fits a 68 GB mesh on the
    host, so the check is the reference's own sharpest self-test
    (algorithms/tests/test_fftpower.py:12-44): the compensated paint of
    a uniform catalog has flat P(k) = shot noise, chi^2/dof < 1 — which
    jointly pins window, FFT normalization and binning at this size."""
    n = int(2e6)
    pos = numpy.random.RandomState(17).uniform(0, 5000., size=(n, 3))
    cat = ArrayCatalog({'Position': pos})
    mesh = cat.to_mesh(Nmesh=2048, BoxSize=5000., dtype='f8',
                       compensated=True, resampler='cic')
    r = FFTPower(mesh, mode='1d', kmax=0.6)
    Pshot = r.attrs['shotnoise']
    valid = r.power['modes'] > 0
    valid[0] = False                      # zero mode is cleared
    P = r.power['power'].real[valid]
    errs = Pshot * (2.0 / r.power['modes'][valid]) ** 0.5
    chisq = (((P - Pshot) / errs) ** 2).sum() / valid.sum()
    assert chisq < 1.5, 'chi2/dof at 2048^3: %g' % chisq
    import torch
    torch.cuda.empty_cache()


@pytest.mark.timeout(900)
@pytest.mark.parametrize('seed', range(6))
def test_random_config_fuzz_gen2(seed):
    """Second-generation fuzz (tests/fuzz_sweep.run_case2): cross
    power, los along any axis, Selection columns, kmax and dk=0 unique
    edges.  64-seed sweep was clean on hardware (r02)."""
    from tests.fuzz_sweep import run_case2
    rel, modes_ok, cfg = run_case2(seed)
    assert modes_ok, cfg
    assert rel < 1e-9, (rel, cfg)


@pytest.mark.timeout(900)
@pytest.mark.parametrize('seed', range(4))
def test_random_recon_fuzz(seed):
    """Reconstruction fuzz (tests/fuzz_sweep.run_case3): random mesh,
    bias, growth rate, smoothing, scheme — vs the oracle.  32-seed
    sweep was clean on hardware (r02)."""
    from tests.fuzz_sweep import run_case3
    rel, _, cfg = run_case3(seed)
    assert rel < 1e-9, (rel, cfg)


@pytest.mark.timeout(900)
@pytest.mark.parametrize('seed', range(4))
def test_random_fkp_fuzz(seed):
    """FKP/survey fuzz (tests/fuzz_sweep.run_case4): random geometry,
    mesh, window, poles, P0, dk — incl. the check that the mesh's own
    ``compensated`` flag is overridden by ConvolvedFFTPower (it always
    compensates, matching the reference).  24-seed hardware sweep was
    clean (r02)."""
    from tests.fuzz_sweep import run_case4
    rel, modes_ok, cfg = run_case4(seed)
    assert modes_ok, cfg
    assert rel < 5e-6, (rel, cfg)       # poles stored as c8


@pytest.mark.timeout(900)
@pytest.mark.parametrize('seed', [0, 1, 2, 5, 8])
def test_random_corr_fuzz(seed):
    """FFTCorr fuzz (tests/fuzz_sweep.run_case5): random mesh, window,
    interlacing, los, poles, cross, rmax and dr=0 unique separations
    (seeds 0/5/8 hit dr0).  64-seed sweep was clean on hardware (r02,
    profiles/r02_fuzz_v5_0.log)."""
    from tests.fuzz_sweep import run_case5
    rel, modes_ok, cfg = run_case5(seed)
    assert modes_ok, cfg
    assert rel < 1e-9, (rel, cfg)


# ---- deferred-x fused FFT + binning path (nbk_fft_x_bin_f64) ----------
# The final x-axis FFT pass runs inside the binning kernel for auto
# power on non-interlaced power-of-two meshes: the finished complex
# field is never materialized.  Parity bar: bit-identical bin counts,
# power to the roundoff tolerances of the other fused-vs-unfused tests.

def _trace_xbin(monkeypatch):
    from nbodykit_amd.algorithms import fftpower as fmod
    # these tests A/B the deferred path against NBK_NO_XBIN themselves —
    # an externally preset value must not leak in
    monkeypatch.delenv('NBK_NO_XBIN', raising=False)
    calls = []
    orig = fmod._project_power_xbin

    def wrapper(*a, **kw):
        calls.append(1)
        return orig(*a, **kw)

    monkeypatch.setattr(fmod, '_project_power_xbin', wrapper)
    return calls


XBIN_CASES = [
    dict(mode='1d'),
    dict(mode='1d', poles=[0, 2, 4]),
    dict(mode='2d', Nmu=5),
    dict(mode='2d', Nmu=3, poles=[0, 2], los=[1, 0, 0]),
    dict(mode='1d', kmin=0.05, kmax=1.0, dk=0.02),
]


@pytest.mark.parametrize('kwargs', XBIN_CASES,
                         ids=['1d', 'poles', '2d', '2dlosx', 'kwin'])
def test_xbin_matches_unfused(kwargs, monkeypatch):
    """Deferred-x path vs the standard fused path (NBK_NO_XBIN=1) on
    the same mesh — the x-FFT element values are bit-identical, so only
    accumulation-order roundoff may differ."""
    cat = UniformCatalog(nbar=3e-3, BoxSize=128., seed=11)
    mesh = cat.to_mesh(Nmesh=128, dtype='f8', compensated=True)
    calls = _trace_xbin(monkeypatch)
    r_x = FFTPower(mesh, **kwargs)
    assert calls, 'deferred-x path did not engage'
    monkeypatch.setenv('NBK_NO_XBIN', '1')
    r_ref = FFTPower(mesh, **kwargs)
    scale = numpy.nanmax(numpy.abs(r_ref.power['power']))
    assert_array_equal(r_x.power['modes'], r_ref.power['modes'])
    assert_allclose(r_x.power['power'], r_ref.power['power'],
                    rtol=1e-10, atol=1e-12 * scale, equal_nan=True)
    assert_allclose(numpy.nan_to_num(r_x.power['k']),
                    numpy.nan_to_num(r_ref.power['k']),
                    rtol=1e-10, atol=1e-12)
    if 'mu' in r_ref.power.variables:
        assert_allclose(numpy.nan_to_num(r_x.power['mu']),
                        numpy.nan_to_num(r_ref.power['mu']),
                        rtol=1e-10, atol=1e-12)
    for ell in kwargs.get('poles', []):
        assert_allclose(r_x.poles['power_%d' % ell],
                        r_ref.poles['power_%d' % ell],
                        rtol=1e-10, atol=1e-12 * scale, equal_nan=True)


@pytest.mark.parametrize('resampler,compensated',
                         [('tsc', True), ('pcs', True), ('cic', False)])
def test_xbin_windows_match_unfused(resampler, compensated, monkeypatch):
    cat = UniformCatalog(nbar=1e-2, BoxSize=64., seed=12)
    mesh = cat.to_mesh(Nmesh=64, dtype='f8', compensated=compensated,
                       resampler=resampler)
    calls = _trace_xbin(monkeypatch)
    r_x = FFTPower(mesh, mode='1d')
    assert calls
    monkeypatch.setenv('NBK_NO_XBIN', '1')
    r_ref = FFTPower(mesh, mode='1d')
    scale = numpy.nanmax(numpy.abs(r_ref.power['power']))
    assert_array_equal(r_x.power['modes'], r_ref.power['modes'])
    assert_allclose(r_x.power['power'], r_ref.power['power'],
                    rtol=1e-10, atol=1e-12 * scale, equal_nan=True)


def test_xbin_through_gather_paint(monkeypatch):
    """Deferred-x composed with the fused paint+z-FFT head: the
    pre-x-pass tensor comes straight out of the gather paint (the C4
    composition).  Compare against the fully unfused pipeline."""
    from nbodykit_amd import set_options
    cat = UniformCatalog(nbar=0.7, BoxSize=64., seed=13)   # ~1.8e5 pts
    calls = _trace_xbin(monkeypatch)
    with set_options(sort_min_n=1024, sort_two_level_min_n=1024,
                     sort_two_level_min_cells=1):
        r_x = FFTPower(cat, mode='1d', Nmesh=64)
    assert calls
    monkeypatch.setenv('NBK_NO_XBIN', '1')
    r_ref = FFTPower(cat, mode='1d', Nmesh=64)
    scale = numpy.nanmax(numpy.abs(r_ref.power['power']))
    assert_array_equal(r_x.power['modes'], r_ref.power['modes'])
    assert_allclose(r_x.power['power'], r_ref.power['power'],
                    rtol=1e-10, atol=1e-12 * scale, equal_nan=True)
    assert_allclose(r_x.attrs['shotnoise'], r_ref.attrs['shotnoise'],
                    rtol=1e-12)


def test_xbin_gates(monkeypatch):
    """Configurations the deferred-x kernel must NOT serve fall back to
    the standard path: interlaced meshes, cross power, dk=0 unique
    edges, non-power-of-two meshes, and NBK_NO_XBIN=1."""
    calls = _trace_xbin(monkeypatch)
    cat = UniformCatalog(nbar=1e-2, BoxSize=64., seed=14)
    cat2 = UniformCatalog(nbar=1e-2, BoxSize=64., seed=15)
    m1 = cat.to_mesh(Nmesh=64, dtype='f8', compensated=True)
    m2 = cat2.to_mesh(Nmesh=64, dtype='f8', compensated=True)
    FFTPower(m1, mode='1d', second=m2)
    assert not calls, 'cross power must not defer'
    FFTPower(m1, mode='1d', dk=0)
    assert not calls, 'dk=0 unique edges must not defer'
    m96 = cat.to_mesh(Nmesh=96, dtype='f8', compensated=True)
    FFTPower(m96, mode='1d')
    assert not calls, 'non-power-of-two mesh must not defer'
    monkeypatch.setenv('NBK_NO_XBIN', '1')
    FFTPower(m1, mode='1d')
    assert not calls, 'NBK_NO_XBIN=1 must not defer'
    monkeypatch.delenv('NBK_NO_XBIN')
    FFTPower(m1, mode='1d')
    assert calls, 'plain auto 1d must defer'


def test_xbin_matches_oracle():
    """Deferred-x FFTPower vs the CPU oracle end-to-end (the engaged
    default path for plain auto power)."""
    nbar, box, nmesh, seed = 1e-2, 128., 128, 21
    cat = UniformCatalog(nbar=nbar, BoxSize=box, seed=seed)
    r = FFTPower(cat, mode='1d', Nmesh=nmesh)
    pos = uniform_positions(nbar, box, seed)
    want = fftpower_oracle(pos, Nmesh=nmesh, BoxSize=box, mode='1d',
                           resampler='cic', compensated=True)
    check_parity(r, want)


@pytest.mark.parametrize('kwargs,mkw', [
    (dict(mode='1d'), dict(resampler='cic')),
    (dict(mode='2d', Nmu=4, poles=[0, 2]), dict(resampler='tsc')),
    (dict(mode='1d'), dict(resampler='pcs', compensated=False)),
], ids=['cic1d', 'tsc2dpoles', 'pcs-nocomp'])
def test_xbin_interlaced_matches_unfused(kwargs, mkw, monkeypatch):
    """Interlaced deferred-x: the kernel FFTs BOTH paints' tiles,
    combines with exp(i k.H/2) and bins; the self-conjugate planes go
    through the host path.  Against the standard interlaced pipeline
    (combine kernel + projection + nbk_power_bin_f64)."""
    cat = UniformCatalog(nbar=1e-2, BoxSize=128., seed=16)
    m = dict(Nmesh=128, dtype='f8', compensated=True, interlaced=True)
    m.update(mkw)
    mesh = cat.to_mesh(**m)
    calls = _trace_xbin(monkeypatch)
    r_x = FFTPower(mesh, **kwargs)
    assert calls, 'interlaced deferred-x did not engage'
    monkeypatch.setenv('NBK_NO_XBIN', '1')
    r_ref = FFTPower(mesh, **kwargs)
    scale = numpy.nanmax(numpy.abs(r_ref.power['power']))
    assert_array_equal(r_x.power['modes'], r_ref.power['modes'])
    assert_allclose(r_x.power['power'], r_ref.power['power'],
                    rtol=1e-10, atol=1e-11 * scale, equal_nan=True)
    for ell in kwargs.get('poles', []):
        assert_allclose(r_x.poles['power_%d' % ell],
                        r_ref.poles['power_%d' % ell],
                        rtol=1e-10, atol=1e-11 * scale, equal_nan=True)


def test_xbin_interlaced_matches_oracle():
    nbar, box, nmesh, seed = 1e-2, 96., 64, 17
    cat = UniformCatalog(nbar=nbar, BoxSize=box, seed=seed)
    mesh = cat.to_mesh(Nmesh=nmesh, dtype='f8', compensated=True,
                       interlaced=True, resampler='tsc')
    r = FFTPower(mesh, mode='1d')
    pos = uniform_positions(nbar, box, seed)
    want = fftpower_oracle(pos, Nmesh=nmesh, BoxSize=box, mode='1d',
                           resampler='tsc', compensated=True,
                           interlaced=True)
    check_parity(r, want)


def test_diagonal_los_parity():
    """Arbitrary (non-axis) unit line-of-sight: mu = k.los/|k| with a
    diagonal los — covers the general dot-product path in the fused
    binning kernels vs the oracle (the reference accepts any unit
    vector, fftpower.py:177-182)."""
    s3 = 1.0 / numpy.sqrt(3.0)
    los = [s3, s3, s3]
    cat = UniformCatalog(nbar=3e-3, BoxSize=100., seed=23)
    r = FFTPower(cat, mode='2d', Nmu=4, Nmesh=64, los=los,
                 poles=[0, 2])
    pos = uniform_positions(3e-3, 100., 23)
    want = fftpower_oracle(pos, Nmesh=64, BoxSize=100., mode='2d',
                           Nmu=4, poles=[0, 2], los=los,
                           resampler='cic', compensated=True)
    check_parity(r, want, poles=[0, 2])


def test_poles_without_monopole():
    """poles=[2, 4] (no explicit 0): the internal ell list prepends 0
    for the normalization but only the requested poles are reported
    (fftpower.py:616-620 semantics)."""
    cat = UniformCatalog(nbar=3e-3, BoxSize=100., seed=24)
    r = FFTPower(cat, mode='1d', Nmesh=64, poles=[2, 4])
    assert 'power_2' in r.poles.variables
    assert 'power_4' in r.poles.variables
    assert 'power_0' not in r.poles.variables
    pos = uniform_positions(3e-3, 100., 24)
    want = fftpower_oracle(pos, Nmesh=64, BoxSize=100., mode='1d',
                           poles=[2, 4], resampler='cic',
                           compensated=True)
    check_parity(r, want, poles=[2, 4])


def test_sort_chunk_invariance(monkeypatch):
    """The counting sorts process particles in fixed-size chunks
    (count-matrix rows); P(k) must be invariant to the chunk size —
    pins the atomic-free placement across chunk boundaries for BOTH
    the pair-bucket and the two-level pipelines."""
    from nbodykit_amd import set_options
    cat = UniformCatalog(nbar=0.7, BoxSize=64., seed=31)   # ~1.8e5 pts
    with set_options(sort_min_n=1024, sort_two_level_min_n=1024,
                     sort_two_level_min_cells=1):
        r_def = FFTPower(cat, mode='1d', Nmesh=64)
        monkeypatch.setenv('NBK_SORT_CHUNK', '10000')
        r_small = FFTPower(cat, mode='1d', Nmesh=64)
        monkeypatch.setenv('NBK_SORT_PAIR', '0')
        r_small_rows = FFTPower(cat, mode='1d', Nmesh=64)
        monkeypatch.setenv('NBK_SORT_CHUNK', '262144')
        r_def_rows = FFTPower(cat, mode='1d', Nmesh=64)
    scale = numpy.nanmax(numpy.abs(r_def.power['power']))
    for r in (r_small, r_small_rows, r_def_rows):
        assert_array_equal(r.power['modes'], r_def.power['modes'])
        assert_allclose(r.power['power'], r_def.power['power'],
                        rtol=1e-11, atol=1e-12 * scale, equal_nan=True)


@pytest.mark.parametrize('resampler', ['cic', 'tsc', 'pcs'])
def test_paint_raw_window_reaches_sort(resampler):
    """paint_raw (the reconstruction driver's bare paint) must pass its
    ACTUAL window to the particle sort: the pair-bucket pipeline's
    duplication range is stencil-dependent, and a CIC-range sort under
    a TSC/PCS paint would silently drop group-boundary deposits
    (regression: the gather branch also carried a stale kernel call
    signature only reachable above the sort thresholds)."""
    import torch
    from nbodykit_amd import set_options
    from nbodykit_amd.source.mesh.catalog import paint_raw
    from nbodykit_amd.pm import ParticleMesh
    pm = ParticleMesh(BoxSize=64., Nmesh=64)
    rng = numpy.random.RandomState(41)
    n = 150000
    pos_t = torch.as_tensor(rng.uniform(0, 64., size=(n, 3))).to('cuda')
    # scatter-path reference (thresholds high: no sort path engages)
    ref = paint_raw(pos_t, pm, resampler=resampler)
    with set_options(sort_min_n=1024, sort_two_level_min_n=1024,
                     sort_two_level_min_cells=1):
        got = paint_raw(pos_t, pm, resampler=resampler)
    a = got.value.cpu().numpy()
    b = ref.value.cpu().numpy()
    assert abs(a.sum() - n) < 1e-5 * n, 'mass not conserved'
    assert_allclose(a, b, rtol=1e-11, atol=1e-11,
                    err_msg='gather/pair paint_raw differs from scatter')


def test_weighted_pair_sort_paint():
    """Weighted catalog through the pair-bucket sort (the mass array is
    duplicated alongside the positions — otherwise untested above the
    thresholds): P(k) must match the scatter-path reference."""
    from nbodykit_amd import set_options
    rng = numpy.random.RandomState(44)
    n = 160000
    cat = ArrayCatalog({'Position': rng.uniform(0, 64., size=(n, 3)),
                        'Weight': rng.exponential(size=n) + 0.1})
    kw = dict(mode='1d', Nmesh=64, BoxSize=64.)
    r_ref = FFTPower(cat, **kw)               # scatter path (small n)
    with set_options(sort_min_n=1024, sort_two_level_min_n=1024,
                     sort_two_level_min_cells=1):
        r_pair = FFTPower(cat, **kw)          # pair sort + gather
    scale = numpy.nanmax(numpy.abs(r_ref.power['power']))
    assert_array_equal(r_pair.power['modes'], r_ref.power['modes'])
    assert_allclose(r_pair.power['power'], r_ref.power['power'],
                    rtol=1e-10, atol=1e-11 * scale, equal_nan=True)
    assert_allclose(r_pair.attrs['shotnoise'], r_ref.attrs['shotnoise'],
                    rtol=1e-12)


@pytest.mark.parametrize('nmesh', [[64, 32, 128], [27, 32, 45],
                                   [32, 45, 64]],
                         ids=['pow2', 'mixed-odd', 'odd-mid'])
def test_noncubic_mesh_parity(nmesh):
    """Non-cubic Nmesh (per-axis dims differ, including MIXED odd/even
    axes — each axis carries its own parity conventions): exercises
    every per-axis code path — freq conventions, the sort geometry
    pickers, tile divisions, per-axis compensation — vs the oracle."""
    nbar, box, seed = 2e-3, 64., 5
    cat = UniformCatalog(nbar=nbar, BoxSize=box, seed=seed)
    r = FFTPower(cat, mode='1d', Nmesh=nmesh)
    pos = uniform_positions(nbar, box, seed)
    want = fftpower_oracle(pos, Nmesh=nmesh, BoxSize=box,
                           mode='1d', resampler='cic', compensated=True)
    check_parity(r, want)


def test_anisotropic_box_parity():
    """Per-axis BoxSize with a non-cubic mesh, TSC + 2d: anisotropic
    k0 = 2 pi / L per axis through paint, compensation and binning."""
    n = 40000
    rng = numpy.random.RandomState(18)
    box = [50., 64., 80.]
    pos = rng.uniform(0, 1, size=(n, 3)) * numpy.asarray(box)
    cat = ArrayCatalog({'Position': pos})
    r = FFTPower(cat, mode='2d', Nmu=4, Nmesh=[32, 64, 48], BoxSize=box)
    want = fftpower_oracle(pos, Nmesh=[32, 64, 48], BoxSize=box,
                           mode='2d', Nmu=4, resampler='cic',
                           compensated=True)
    check_parity(r, want)


def test_fftcorr_unique_edges():
    """FFTCorr with dr=0: unique configuration-space separations as bin
    edges (reference test_fftcorr.py:28-36)."""
    from nbodykit_amd.lab import FFTCorr
    from oracle import fftcorr_oracle
    n = 20000
    pos = numpy.random.RandomState(10).uniform(0, 100., size=(n, 3))
    cat = ArrayCatalog({'Position': pos})
    r = FFTCorr(cat, mode='1d', Nmesh=16, BoxSize=100., dr=0)
    want = fftcorr_oracle(pos, Nmesh=16, BoxSize=100., mode='1d',
                          resampler='cic', compensated=True, dr=0)
    assert_array_equal(r.corr['modes'], want['modes'])
    got = r.corr['corr']
    ref = want['corr']
    ok = numpy.isfinite(ref) & (numpy.abs(ref) > 1e-12)
    rel = numpy.abs(got[ok] - ref[ok]) / numpy.abs(ref[ok])
    assert rel.max() < 1e-9, 'dr=0 FFTCorr parity: %g' % rel.max()
