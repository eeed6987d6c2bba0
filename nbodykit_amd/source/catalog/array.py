"""
ArrayCatalog — wrap a structured array / dict of arrays as a catalog
(reference nbodykit/source/catalog/array.py; handy for feeding explicit
positions into the pipeline and for tests).
"""
import numpy

from nbodykit_amd import CurrentMPIComm
from nbodykit_amd.base.catalog import CatalogSource


class ArrayCatalog(CatalogSource):

    @CurrentMPIComm.enable
    def __init__(self, data, comm=None, **kwargs):
        self.comm = comm
        if isinstance(data, numpy.ndarray):
            if data.dtype.names is None:
                raise ValueError("input to ArrayCatalog must be a "
                                 "structured array or a dict of arrays")
            cols = {name: data[name] for name in data.dtype.names}
        elif isinstance(data, dict):
            cols = dict(data)
        else:
            raise ValueError("input to ArrayCatalog must be a structured "
                             "array or a dict of arrays")

        sizes = set(len(v) for v in cols.values())
        if len(sizes) > 1:
            raise ValueError("columns of unequal length: %s" % sizes)
        self._size = sizes.pop() if sizes else 0

        CatalogSource.__init__(self, comm=comm)
        self.attrs.update(kwargs)
        for name, v in cols.items():
            self[name] = numpy.asarray(v)
