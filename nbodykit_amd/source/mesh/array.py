"""
ArrayMesh — a MeshSource initialized from an in-memory numpy array
(reference nbodykit/source/mesh/array.py:8-81).  A complex input array
is transformed to configuration space on the host first (reference
:34-37); the local slab is uploaded on compute.
"""
import logging

import numpy

from nbodykit_amd import CurrentMPIComm
from nbodykit_amd.base.mesh import MeshSource
from nbodykit_amd.pm import RealField


class ArrayMesh(MeshSource):
    logger = logging.getLogger('ArrayMesh')

    def __repr__(self):
        return "ArrayMesh()"

    @CurrentMPIComm.enable
    def __init__(self, array, BoxSize, comm=None, root=0, **kwargs):
        if comm.rank == root:
            array = numpy.array(array)
            if array.dtype.kind == 'c':
                # transform to real for the correct shape (reference
                # :34-37; their irfftn is unnormalized-forward so the
                # size factor restores the amplitude)
                array = numpy.fft.irfftn(array, axes=range(array.ndim))
                array[...] *= numpy.prod(array.shape)
            shape = array.shape
        else:
            array, shape = None, None
        shape = comm.bcast(shape, root=root)
        if len(shape) != 3:
            raise ValueError("ArrayMesh accepts 3d arrays (2d meshes are "
                             "not supported by this build)")
        # every rank needs its slab: broadcast (the reference instead
        # scatters via pmesh unravel; the array is root-hosted either way)
        array = comm.bcast(array, root=root)

        MeshSource.__init__(self, comm, shape, BoxSize, 'f8')
        self.attrs.update(kwargs)
        self._array = numpy.ascontiguousarray(array, dtype='f8')

    def to_real_field(self, out=None, normalize=True):
        import torch
        pm = self.pm
        local = self._array[pm.x_start:pm.x_start + pm.nx_local]
        t = torch.as_tensor(local).to('cuda')
        f = RealField(pm, tensor=t.contiguous())
        f.attrs = dict(self.attrs)
        return f
