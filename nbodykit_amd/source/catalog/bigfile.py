"""
BigFileCatalog — a CatalogSource reading a bigfile directory (reference
nbodykit/io/bigfile.py:16-120 + source/catalog/file.py:234: the factory
produces a FileCatalogBase over io.BigFile; here the io layer and the
catalog are one class since columns are numpy-backed).  Rows are split
evenly across ranks (FileCatalogBase partitioning,
source/catalog/file.py:67-90).
"""
import json
import logging
from fnmatch import fnmatch

import numpy

from nbodykit_amd.base.catalog import CatalogSource
from nbodykit_amd.io.bigfile_format import BigFile, Dataset
from nbodykit_amd.utils import JSONDecoder


class Automatic(object):
    pass


class BigFileCatalog(CatalogSource):
    logger = logging.getLogger('BigFileCatalog')

    def __repr__(self):
        return "BigFileCatalog(size=%d, file=%r)" % (self.size, self.path)

    def __init__(self, path, exclude=None, header=Automatic, dataset='./',
                 comm=None, attrs=None):
        if not dataset.endswith('/'):
            dataset = dataset + '/'

        self.path = path
        self.dataset = dataset

        ff = BigFile(path, comm=comm)
        columns = ff[self.dataset].blocks
        headers = self._find_headers(header, dataset, ff)

        if exclude is None:
            exclude = headers
        if not isinstance(exclude, (list, tuple)):
            exclude = [exclude]
        columns = [c for c in set(columns)
                   if not any(fnmatch(c, e) for e in exclude)]

        ds = Dataset(ff[self.dataset], columns)

        file_attrs = {}
        for h in headers:
            try:
                battrs = ff[h].attrs
            except KeyError:
                continue
            for k in battrs.keys():
                v = battrs[k]
                if isinstance(v, str) and v.startswith('json://'):
                    file_attrs[k] = json.loads(v[7:], cls=JSONDecoder)
                else:
                    file_attrs[k] = numpy.array(v, copy=True)

        # even row partition across ranks (file.py:67-90)
        if comm is None:
            from nbodykit_amd import CurrentMPIComm
            comm = CurrentMPIComm.get()
        size = ds.size
        start = size * comm.rank // comm.size
        end = size * (comm.rank + 1) // comm.size
        self._size = end - start

        CatalogSource.__init__(self, comm=comm)
        self.attrs.update(file_attrs)
        if attrs is not None:
            self.attrs.update(attrs)

        for col in ds.columns:
            self._overrides[col] = ds[col][start:end]

    def _find_headers(self, header, dataset, ff):
        """Header candidates: Header/header/. + the dataset's own '.'
        (reference io/bigfile.py:97-120)."""
        if header is Automatic:
            header = ['Header', 'header', '.']
        if not isinstance(header, (tuple, list)):
            header = [header]
        blocks = ff.blocks
        r = [h for h in header if h in blocks]
        ds_head = dataset.strip('/') + '/.'
        if ds_head not in r:
            r.append(ds_head)
        return r
