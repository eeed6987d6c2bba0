"""
File-backed catalogs (reference nbodykit/io/binary.py, io/csv.py +
source/catalog/file.py factories): the io layer and the catalog are one
class here since columns are numpy-backed; rows are split evenly across
ranks like the reference's FileCatalogBase (file.py:67-90).
"""
import glob as _glob
import logging
import os

import numpy

from nbodykit_amd.base.catalog import CatalogSource


def _partition(csize, comm):
    start = csize * comm.rank // comm.size
    end = csize * (comm.rank + 1) // comm.size
    return start, end


def _expand_paths(path):
    """A path, a glob pattern, or a list of either -> sorted file list
    (the reference's FileStack, io/stack.py: multiple physical files
    form one logical catalog)."""
    if isinstance(path, (list, tuple)):
        out = []
        for p in path:
            out.extend(_expand_paths(p))
        return out
    if any(ch in path for ch in '*?['):
        hits = sorted(_glob.glob(path))
        if not hits:
            raise FileNotFoundError("no files match %r" % path)
        return hits
    return [path]


class BinaryCatalog(CatalogSource):
    """Catalog from a COLUMN-MAJOR binary file (reference
    io/binary.py:33-144: each column stored contiguously, optional
    per-column byte offsets, fixed header)."""
    logger = logging.getLogger('BinaryCatalog')

    def __repr__(self):
        return "BinaryCatalog(size=%d, file=%r)" % (self.size, self.path)

    def __init__(self, path, dtype, offsets=None, header_size=0,
                 size=None, comm=None, attrs=None):
        paths = _expand_paths(path)
        if len(paths) > 1:
            if offsets is not None or size is not None:
                raise ValueError("offsets/size only apply to a single "
                                 "file")
            parts = [BinaryCatalog(p, dtype, header_size=header_size,
                                   comm=comm, attrs=attrs)
                     for p in paths]
            from nbodykit_amd.transform import ConcatenateSources
            stacked = ConcatenateSources(*parts)
            self.path = paths
            self._size = stacked.size
            CatalogSource.__init__(self, comm=stacked.comm)
            self._overrides.update(stacked._overrides)
            self._attrs = dict(stacked.attrs)
            return
        self.path = path = paths[0]
        dtype = numpy.dtype(dtype)
        if dtype.names is None:
            raise ValueError("input dtype should be structured (a list "
                             "of (name, dtype) tuples)")

        if size is None:
            nbytes = os.path.getsize(path) - header_size
            size, rem = divmod(nbytes, dtype.itemsize)
            if rem != 0:
                raise ValueError("byte size mismatch -- fractional rows "
                                 "found")
        size = int(size)

        if offsets is None:
            offsets = {}
            offset = header_size
            for col in dtype.names:
                offsets[col] = offset
                offset += dtype[col].itemsize * size
        else:
            offsets = dict(offsets)
            missing = [c for c in dtype.names if c not in offsets]
            if missing:
                raise ValueError("`offsets` must contain every column; "
                                 "missing %s" % missing)

        from nbodykit_amd import CurrentMPIComm
        comm = comm if comm is not None else CurrentMPIComm.get()
        start, end = _partition(size, comm)
        self._size = end - start

        CatalogSource.__init__(self, comm=comm)
        if attrs is not None:
            self.attrs.update(attrs)

        with open(path, 'rb') as ff:
            for col in dtype.names:
                sub = dtype[col]
                base = sub.base if sub.shape else sub
                nmemb = int(numpy.prod(sub.shape)) if sub.shape else 1
                ff.seek(offsets[col] + start * base.itemsize * nmemb)
                data = numpy.fromfile(ff, dtype=base,
                                      count=(end - start) * nmemb)
                if sub.shape:
                    data = data.reshape((end - start,) + sub.shape)
                self._overrides[col] = data


class CSVCatalog(CatalogSource):
    """Catalog from a CSV (or whitespace-delimited) text file via
    pandas (reference io/csv.py:140-290: no header row, column ``names``
    required, ``delim_whitespace`` default True, per-column ``dtype``
    overrides, ``usecols`` selection).  The reference partitions the
    file into dask blocks; here pandas reads it once and rows are split
    across ranks."""
    logger = logging.getLogger('CSVCatalog')

    def __repr__(self):
        return "CSVCatalog(size=%d, file=%r)" % (self.size, self.path)

    def __init__(self, path, names, blocksize=None, dtype={},
                 usecols=None, delim_whitespace=True, comm=None,
                 attrs=None, **config):
        import pandas as pd
        paths = _expand_paths(path)
        if len(paths) > 1:
            parts = [CSVCatalog(p, names, dtype=dtype, usecols=usecols,
                                delim_whitespace=delim_whitespace,
                                comm=comm, attrs=attrs, **config)
                     for p in paths]
            from nbodykit_amd.transform import ConcatenateSources
            stacked = ConcatenateSources(*parts)
            self.path = paths
            self._size = stacked.size
            CatalogSource.__init__(self, comm=stacked.comm)
            self._overrides.update(stacked._overrides)
            self._attrs = dict(stacked.attrs)
            return
        self.path = path = paths[0]

        if isinstance(dtype, numpy.dtype) or numpy.isscalar(dtype) \
                or isinstance(dtype, (str, type)):
            dtype = {col: dtype for col in names}

        kws = dict(config)
        kws['header'] = None
        kws['names'] = names
        if usecols is not None:
            kws['usecols'] = usecols
        if delim_whitespace:
            kws['sep'] = r'\s+'
        frame = pd.read_csv(path, **kws)

        cols = usecols if usecols is not None else names
        size = len(frame)

        from nbodykit_amd import CurrentMPIComm
        comm = comm if comm is not None else CurrentMPIComm.get()
        start, end = _partition(size, comm)
        self._size = end - start

        CatalogSource.__init__(self, comm=comm)
        if attrs is not None:
            self.attrs.update(attrs)

        for col in cols:
            data = frame[col].to_numpy()[start:end]
            want = dtype.get(col, 'f8')
            self._overrides[col] = data.astype(want)
