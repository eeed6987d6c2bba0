"""
BAO reconstruction flow (reference cookbook: FFTRecon): displace
galaxies and randoms by the smoothed Zel'dovich estimate and measure
the pre/post power.  Run on an MI355X: python examples/recon_demo.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))

import numpy

from nbodykit_amd.lab import (LogNormalCatalog, LinearPower, FFTPower,
                              FFTRecon, UniformCatalog)
from nbodykit_amd.cosmology import Planck15

Plin = LinearPower(Planck15, redshift=0.55, transfer='EisensteinHu')
data = LogNormalCatalog(Plin=Plin, nbar=1e-4, BoxSize=500., Nmesh=128,
                        bias=2.0, seed=42)
ran = UniformCatalog(nbar=5e-4, BoxSize=500., seed=7)

recon = FFTRecon(data=data, ran=ran, Nmesh=128, bias=2.0, R=20.,
                 scheme='LGS')

r_pre = FFTPower(data, mode='1d', Nmesh=128)
r_post = FFTPower(recon, mode='1d', Nmesh=128)

k = r_pre.power['k']
print('k head:', k[1:4])
print('P pre :', numpy.real(r_pre.power['power'][1:4]))
print('P post:', numpy.real(r_post.power['power'][1:4]))
r_post.save('recon_demo.json')
print('saved recon_demo.json')
