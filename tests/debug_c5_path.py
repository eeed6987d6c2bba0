"""Trace which compute paths a C5-shaped cross FFTPower takes (GPU)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy
import torch

import bench
from nbodykit_amd import set_options
from nbodykit_amd.source.mesh import catalog as cmod
from nbodykit_amd.algorithms import fftpower as fmod
from nbodykit_amd.lab import FFTPower
from nbodykit_amd.source.catalog.device import DeviceArrayCatalog

orig_tcf = cmod.CatalogMesh.to_complex_field
def tcf(self, out=None):
    r = orig_tcf(self, out=out)
    print('[trace] to_complex_field ->',
          'NotImplemented' if r is NotImplemented else 'fused', flush=True)
    return r
cmod.CatalogMesh.to_complex_field = tcf

orig_prep = cmod._prepare_particles
def prep(pos_t, mass_t, pm, force_rowtab=False, **kw):
    r = orig_prep(pos_t, mass_t, pm, force_rowtab, **kw)
    print('[trace] prepare n=%d force=%s table=%s sorted=%s'
          % (len(pos_t), force_rowtab,
             'pair' if isinstance(r[3], tuple) else r[3] is not None,
             r[2]), flush=True)
    return r
cmod._prepare_particles = prep

orig_fuse = fmod._fuse_info
def finfo(mesh):
    r = orig_fuse(mesh)
    print('[trace] _fuse_info ->', r, flush=True)
    return r
fmod._fuse_info = finfo

n = int(5e8)
nmesh = 1024
box = 5000.
pos = bench.gen_lognormal(n, nmesh, box, 0, 1, seed=42)
pos2 = bench.gen_lognormal(n, nmesh, box, 0, 1, seed=43)
ba = numpy.array([box] * 3)
cat = DeviceArrayCatalog({'Position': pos}, BoxSize=ba)
cat2 = DeviceArrayCatalog({'Position': pos2}, BoxSize=ba)
with set_options(paint_chunk_size=1 << 30):
    m1 = cat.to_mesh(Nmesh=nmesh, dtype='f8', compensated=True,
                     resampler='cic')
    m2 = cat2.to_mesh(Nmesh=nmesh, dtype='f8', compensated=True,
                      resampler='cic')
    r = FFTPower(m1, mode='2d', Nmu=5, second=m2)
print('[trace] done; power[1] =', r.power['power'][1][:2], flush=True)
