"""
The reference's survey-data flow (nbodykit docs, cookbook/convpower):
sky coordinates -> Cartesian positions -> FKP catalog -> multipoles.
Run on an MI355X: python examples/survey_convpower_demo.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))

import numpy

from nbodykit_amd.lab import (transform, ArrayCatalog, FKPCatalog,
                              ConvolvedFFTPower, RedshiftHistogram)
from nbodykit_amd.cosmology import Planck15

rng = numpy.random.RandomState(42)


def make_survey(n):
    ra = rng.uniform(120., 150., n)
    dec = rng.uniform(-5., 25., n)
    z = rng.uniform(0.4, 0.7, n)
    pos = transform.SkyToCartesian(ra, dec, z, Planck15)
    return ArrayCatalog({'Position': pos, 'NZ': numpy.full(n, 3e-4)})


data = make_survey(50000)
randoms = make_survey(500000)

# measure n(z) from the randoms and refresh the NZ columns, like the
# reference's convpower cookbook (fsky of the 30x30 deg patch)
fsky = (numpy.deg2rad(30.) * (numpy.sin(numpy.deg2rad(25.))
                              - numpy.sin(numpy.deg2rad(-5.)))) \
    / (4 * numpy.pi)
randoms['Redshift'] = rng.uniform(0.4, 0.7, randoms.size)
data['Redshift'] = rng.uniform(0.4, 0.7, data.size)
nz = RedshiftHistogram(randoms, fsky, Planck15, redshift='Redshift')
alpha = 1.0 * data.csize / randoms.csize
randoms['NZ'] = nz.interpolate(numpy.asarray(randoms['Redshift'])) * alpha
data['NZ'] = nz.interpolate(numpy.asarray(data['Redshift'])) * alpha

fkp = FKPCatalog(data, randoms, P0=1e4)
mesh = fkp.to_mesh(Nmesh=128, dtype='f8', compensated=True)

r = ConvolvedFFTPower(mesh, poles=[0, 2, 4], dk=0.01)
print('alpha =', r.attrs['alpha'])
print('shotnoise =', r.attrs['shotnoise'])
print('P0 head:', numpy.real(r.poles['power_0'][1:5])
      - r.attrs['shotnoise'])
print('P2 head:', numpy.real(r.poles['power_2'][1:5]))

pkmu = r.to_pkmu(numpy.linspace(0, 1, 4), max_ell=4)
print('P(k,mu) wedges shape:', pkmu['power'].shape)
r.save('convpower_demo.json')
print('saved convpower_demo.json')
