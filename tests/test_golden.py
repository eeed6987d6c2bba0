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
