"""
Oracle-vs-golden regression pins: the committed fixtures under
tests/golden/ were produced by oracle/make_golden.py; re-running the
oracle here must reproduce them bitwise-close.  (These fixtures are also
the parity anchor on the GPU box, where /root/reference is absent.)
"""
import glob
import json
import os

import numpy
import pytest
from numpy.testing import assert_allclose

from nbodykit_amd.utils import JSONDecoder

HERE = os.path.dirname(os.path.abspath(__file__))
GOLDEN = sorted(glob.glob(os.path.join(HERE, 'golden',
                                       'oracle_fftpower_*.json')))


def load_golden(path):
    with open(path) as ff:
        return json.load(ff, cls=JSONDecoder)


def rerun(g):
    from oracle import fftpower_oracle
    from oracle.make_golden import make_positions
    cfg = g['config']
    pos = make_positions((cfg['pos'][0], cfg['pos'][1]))
    second = None
    if cfg.get('second'):
        second = make_positions((cfg['second'][0], cfg['second'][1]))
    return fftpower_oracle(pos, second_position=second, **cfg['run'])


@pytest.mark.parametrize('path', GOLDEN, ids=[os.path.basename(p)
                                              for p in GOLDEN])
def test_oracle_matches_golden(path):
    g = load_golden(path)
    r = rerun(g)
    assert_allclose(r['kedges'], g['kedges'], rtol=1e-12)
    assert_allclose(numpy.nan_to_num(r['power']),
                    numpy.nan_to_num(g['power']), rtol=1e-10, atol=1e-8)
    assert numpy.array_equal(r['modes'], g['modes'])
    assert_allclose(r['attrs']['shotnoise'], g['shotnoise'], rtol=1e-12)
    for key in [k for k in g if k.startswith('power_')]:
        ell = int(key.split('_')[1])
        assert_allclose(numpy.nan_to_num(r['poles'][ell]),
                        numpy.nan_to_num(g[key]), rtol=1e-10, atol=1e-8)


def test_golden_files_exist():
    assert len(GOLDEN) >= 5


def test_oracle_reproduces_fftcorr_golden():
    """Drift detector: the oracle still reproduces the committed
    FFTCorr vector (the GPU parity test compares the product against
    the live oracle)."""
    import json
    import os
    import numpy
    from numpy.testing import assert_allclose, assert_array_equal
    from nbodykit_amd.utils import JSONDecoder
    from oracle import fftcorr_oracle
    from oracle.make_golden import uniform_positions

    path = os.path.join(os.path.dirname(__file__), 'golden',
                        'oracle_fftcorr_uniform_cic_1d.json')
    with open(path) as ff:
        want = json.load(ff, cls=JSONDecoder)
    c = want['input']
    pos = uniform_positions(c['nbar'], c['BoxSize'], seed=c['seed'])
    got = fftcorr_oracle(pos, Nmesh=c['Nmesh'], BoxSize=c['BoxSize'],
                         mode=c['mode'], resampler=c['resampler'],
                         compensated=True)
    assert_array_equal(got['modes'], want['modes'])
    assert_allclose(got['corr'], want['corr'], rtol=1e-12,
                    equal_nan=True)


def test_oracle_reproduces_convpower_golden():
    import json
    import os
    import numpy
    from numpy.testing import assert_allclose, assert_array_equal
    from nbodykit_amd.utils import JSONDecoder
    from oracle.convpower import convpower_oracle

    path = os.path.join(os.path.dirname(__file__), 'golden',
                        'oracle_convpower_poles02.json')
    with open(path) as ff:
        want = json.load(ff, cls=JSONDecoder)
    c = want['input']
    rng = numpy.random.RandomState(c['seed'])
    lo = numpy.array([900., 900., 900.])
    span = numpy.array([200., 200., 200.])
    dpos = lo + rng.uniform(0., 1., size=(c['nd'], 3)) * span
    rpos = lo + rng.uniform(0., 1., size=(c['nr'], 3)) * span
    nbar = numpy.full(c['nd'], c['nd'] / span.prod())
    nbarr = numpy.full(c['nr'], c['nd'] / span.prod())
    got = convpower_oracle(dpos, rpos, c['poles'], Nmesh=c['Nmesh'],
                           BoxSize=c['BoxSize'],
                           BoxCenter=c['BoxCenter'],
                           nbar_data=nbar, nbar_ran=nbarr,
                           compensated=True, dk=c['dk'])
    assert_array_equal(got['modes'], want['modes'])
    assert_allclose(got['attrs']['alpha'], want['alpha'], rtol=1e-12)
    assert_allclose(got['attrs']['shotnoise'], want['shotnoise'],
                    rtol=1e-12)
    for ell in c['poles']:
        assert_allclose(got['power_%d' % ell], want['power_%d' % ell],
                        rtol=1e-6, atol=1e-8, equal_nan=True)
