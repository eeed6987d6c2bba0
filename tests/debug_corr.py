"""Reproduce a failing FFTCorr fuzz seed with detail (GPU)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy

from nbodykit_amd.lab import ArrayCatalog, FFTCorr
from oracle.fftpower import fftcorr_oracle

seed = int(sys.argv[1]) if len(sys.argv) > 1 else 63
rng = numpy.random.RandomState(5000 + seed)
nmesh = int(rng.choice([32, 48, 64, 96, 128, 160]))
box = float(rng.uniform(100., 1000.))
window = str(rng.choice(['cic', 'tsc', 'pcs']))
interlaced = bool(rng.randint(2))
compensated = bool(rng.randint(2))
mode = str(rng.choice(['1d', '2d']))
Nmu = int(rng.choice([3, 5]))
poles = [0, 2] if rng.randint(2) else []
kmin = float(rng.choice([0.0, 0.02]))
n = int(rng.randint(20000, 120000))
pos = rng.uniform(0, box, size=(n, 3))
weight = rng.uniform(0.5, 2.0, size=n) if rng.randint(2) else None
print('cfg', nmesh, box, window, 'interl', interlaced, 'comp',
      compensated, mode, 'poles', poles, 'n', n,
      'weighted', weight is not None, flush=True)

cat = ArrayCatalog({'Position': pos} if weight is None
                   else {'Position': pos, 'Weight': weight})
mesh = cat.to_mesh(Nmesh=nmesh, BoxSize=box, dtype='f8',
                   compensated=compensated, resampler=window,
                   interlaced=interlaced)
kw = dict(mode=mode, poles=poles)
if mode == '2d':
    kw['Nmu'] = Nmu
r = FFTCorr(mesh, **kw)
want = fftcorr_oracle(pos, weight=weight, Nmesh=nmesh, BoxSize=box,
                      resampler=window, compensated=compensated,
                      interlaced=interlaced, Nmu=Nmu, poles=poles,
                      mode=mode)
gm = numpy.asarray(r.corr['modes'])
wm = numpy.asarray(want['modes'])
print('shapes', gm.shape, wm.shape, flush=True)
print('redges prod', r.corr.edges['r'][:4], '...', r.corr.edges['r'][-2:])
print('redges orac', want['redges'][:4], '...', want['redges'][-2:])
if gm.shape == wm.shape:
    d = numpy.flatnonzero(numpy.ravel(gm) != numpy.ravel(wm))
    print('mode-diff flat idx', d[:20])
    print('gpu ', numpy.ravel(gm)[d[:10]])
    print('cpu ', numpy.ravel(wm)[d[:10]])
    gc = numpy.nan_to_num(numpy.ravel(numpy.asarray(r.corr['corr']).real))
    wc = numpy.nan_to_num(numpy.ravel(numpy.asarray(want['corr']).real))
    rd = numpy.abs(gc - wc)
    i = numpy.argsort(-rd)[:10]
    print('biggest corr diffs at', i, 'gpu', gc[i], 'cpu', wc[i])
