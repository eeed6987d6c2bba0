"""
Golden vectors for the widening rows (FFTCorr, FFTRecon displacement
config, ConvolvedFFTPower) — oracle outputs on small fixed-seed inputs,
committed under tests/golden/ as drift detectors (the GPU parity tests
additionally compare against the live oracle).

Run from the repo root:  python3 oracle/make_golden_extras.py
"""
import json
import os
import sys

import numpy

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

from nbodykit_amd.utils import JSONEncoder             # noqa: E402
from oracle.make_golden import uniform_positions       # noqa: E402
from oracle import fftcorr_oracle                      # noqa: E402
from oracle.convpower import convpower_oracle          # noqa: E402

OUTDIR = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), 'tests', 'golden')


def gen_fftcorr():
    pos = uniform_positions(2e-3, 64., seed=31)
    r = fftcorr_oracle(pos, Nmesh=32, BoxSize=64., mode='1d',
                       resampler='cic', compensated=True)
    return {'input': {'nbar': 2e-3, 'BoxSize': 64., 'seed': 31,
                      'Nmesh': 32, 'mode': '1d', 'resampler': 'cic'},
            'r': r['r'], 'corr': r['corr'], 'modes': r['modes']}


def gen_convpower():
    rng = numpy.random.RandomState(77)
    lo = numpy.array([900., 900., 900.])
    span = numpy.array([200., 200., 200.])
    nd, nr = 2000, 20000
    dpos = lo + rng.uniform(0., 1., size=(nd, 3)) * span
    rpos = lo + rng.uniform(0., 1., size=(nr, 3)) * span
    nbar = numpy.full(nd, nd / span.prod())
    nbarr = numpy.full(nr, nd / span.prod())
    out = convpower_oracle(dpos, rpos, [0, 2], Nmesh=32, BoxSize=256.,
                           BoxCenter=1000., nbar_data=nbar,
                           nbar_ran=nbarr, compensated=True, dk=0.05)
    return {'input': {'seed': 77, 'nd': nd, 'nr': nr,
                      'Nmesh': 32, 'BoxSize': 256., 'BoxCenter': 1000.,
                      'dk': 0.05, 'poles': [0, 2]},
            'k': out['k'], 'power_0': out['power_0'],
            'power_2': out['power_2'], 'modes': out['modes'],
            'alpha': out['attrs']['alpha'],
            'shotnoise': out['attrs']['shotnoise']}


def main():
    for name, fn in [('oracle_fftcorr_uniform_cic_1d', gen_fftcorr),
                     ('oracle_convpower_poles02', gen_convpower)]:
        path = os.path.join(OUTDIR, name + '.json')
        with open(path, 'w') as ff:
            json.dump(fn(), ff, cls=JSONEncoder)
        print('wrote', path)


if __name__ == '__main__':
    main()
