"""
BigFileMesh — load a mesh saved by ``MeshSource.save`` (reference
nbodykit/source/mesh/bigfile.py:16-137).
"""
import json
import logging

import numpy

from nbodykit_amd import CurrentMPIComm
from nbodykit_amd.base.mesh import MeshSource
from nbodykit_amd.io.bigfile_format import BigFile
from nbodykit_amd.pm import RealField, ComplexField
from nbodykit_amd.utils import JSONDecoder


class BigFileMesh(MeshSource):
    logger = logging.getLogger('BigFileMesh')

    def __repr__(self):
        import os
        return "BigFileMesh(file=%s)" % os.path.basename(self.path)

    @CurrentMPIComm.enable
    def __init__(self, path, dataset, comm=None, **kwargs):
        self.path = path
        self.dataset = dataset

        self.attrs.update(kwargs)
        ff = BigFile(path, comm=comm)[dataset]
        for key in ff.attrs:
            v = ff.attrs[key]
            if isinstance(v, str) and v.startswith('json://'):
                self.attrs[key] = json.loads(v[7:], cls=JSONDecoder)
            else:
                self.attrs[key] = numpy.squeeze(v)

        # fourier or configuration space, and the compute dtype
        # (reference :55-66)
        if ff.dtype.kind == 'c':
            self.isfourier = True
        else:
            self.isfourier = False
        dtype = 'f8' if ff.dtype.itemsize in (8, 16) else 'f4'

        if 'ndarray.shape' not in self.attrs:
            raise ValueError("`ndarray.shape` should be stored in the "
                             "Bigfile `attrs` to determine `Nmesh`")
        if 'Nmesh' not in self.attrs:
            raise ValueError("`Nmesh` should be stored in the Bigfile "
                             "`attrs` to determine `Nmesh`")

        MeshSource.__init__(self, comm, self.attrs['Nmesh'],
                            self.attrs['BoxSize'], dtype)

    def _load(self, field):
        import torch
        comm = self.comm
        ds = BigFile(self.path, comm=comm)[self.dataset]
        size = int(numpy.prod(field.value.shape))
        start = sum(comm.allgather(size)[:comm.rank])
        flat = ds[start:start + size]
        t = torch.from_numpy(
            numpy.ascontiguousarray(flat)).to('cuda')
        field.value.copy_(t.view(field.value.shape))
        field.attrs = dict(self.attrs)
        return field

    def to_real_field(self, out=None, normalize=True):
        """The RealField stored on disk (must have been saved with
        mode='real')."""
        if self.isfourier:
            return NotImplemented
        return self._load(RealField(self.pm))

    def to_complex_field(self, out=None):
        if not self.isfourier:
            return NotImplemented
        return self._load(ComplexField(self.pm))
