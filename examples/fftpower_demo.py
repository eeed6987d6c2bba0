"""
The reference's periodic-box cookbook flow (nbodykit docs,
cookbook/fftpower), unchanged except for the import line.
Run on an MI355X: python examples/fftpower_demo.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))

import numpy

from nbodykit_amd.lab import (LogNormalCatalog, LinearPower, FFTPower,
                              FFTCorr, LinearMesh)
from nbodykit_amd.cosmology import Planck15

Plin = LinearPower(Planck15, redshift=0.55, transfer='EisensteinHu')
cat = LogNormalCatalog(Plin=Plin, nbar=3e-4, BoxSize=1380., Nmesh=256,
                       bias=2.0, seed=42)

# P(k, mu) wedges + multipoles in one pass
r = FFTPower(cat, mode='2d', Nmesh=256, Nmu=5, poles=[0, 2, 4])
pk = r.power
print('shotnoise =', r.attrs['shotnoise'])
for i in range(pk.shape[1]):
    sl = pk[:, i]
    print('mu =', pk.coords['mu'][i], ' P(k) head:',
          numpy.real(sl['power'][1:4] - r.attrs['shotnoise']))

# the quadrupole
p2 = r.poles['power_2']
print('P2 head:', numpy.real(p2[1:4]))

# correlation function of the same catalog
xi = FFTCorr(cat, mode='1d', Nmesh=256)
print('xi(r) head:', numpy.real(xi.corr['corr'][1:4]))

# a Gaussian realization directly from P(k)
mesh = LinearMesh(Plin, BoxSize=1380., Nmesh=128, seed=7)
rl = FFTPower(mesh, mode='1d')
print('LinearMesh P(k) head:', numpy.real(rl.power['power'][1:4]))

r.save('fftpower_demo.json')       # nbodykit-compatible JSON
print('saved fftpower_demo.json')
