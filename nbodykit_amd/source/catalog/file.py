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


GADGET_HEADER_DTYPE = [
    ('Npart', ('u4', 6)),
    ('Massarr', ('f8', 6)),
    ('Time', 'f8'),
    ('Redshift', 'f8'),
    ('FlagSfr', 'i4'),
    ('FlagFeedback', 'i4'),
    ('Nall', ('u4', 6)),
    ('FlagCooling', 'i4'),
    ('NumFiles', 'i4'),
    ('BoxSize', 'f8'),
    ('Omega0', 'f8'),
    ('OmegaLambda', 'f8'),
    ('HubbleParam', 'f8'),
    ('FlagAge', 'i4'),
    ('FlagMetals', 'i4'),
    ('NallHW', ('u4', 6)),
    ('flag_entr_ics', 'i4'),
]

GADGET_COLUMN_DEFS = [
    ('Position', ('auto', 3), 'all'),
    ('GadgetVelocity', ('auto', 3), 'all'),
    ('ID', 'auto', 'all'),
    ('Mass', 'auto', None),
    ('InternalEnergy', 'auto', (0,)),
    ('Density', 'auto', (0,)),
    ('SmoothingLength', 'auto', (0,)),
]


class Gadget1Catalog(CatalogSource):
    """Catalog from a classic Gadget-1 (F77-unformatted) snapshot
    (reference io/gadget.py:6-218): 256-byte header + per-column blocks
    bracketed by i4 sizes; the float width of each block is inferred
    from its size marker; particles of ``ptype`` are selected and a
    constant ``Massarr`` mass is broadcast when the Mass block is
    absent.  Header fields land in ``attrs``."""
    logger = logging.getLogger('Gadget1Catalog')

    def __repr__(self):
        return "Gadget1Catalog(size=%d, file=%r)" % (self.size, self.path)

    def __init__(self, path, columndefs=GADGET_COLUMN_DEFS, ptype=1,
                 hdtype=GADGET_HEADER_DTYPE, comm=None, attrs=None):
        self.path = path
        hdtype = numpy.dtype(hdtype)
        pad = numpy.dtype([('header', hdtype),
                           ('padding', ('u1', 256 - hdtype.itemsize))])

        with open(path, 'rb') as ff:
            hsize = numpy.fromfile(ff, dtype='i4', count=1)[0]
            if hsize != 256:
                raise IOError("header block size %d != 256 — not a "
                              "Gadget-1 snapshot" % hsize)
            header = numpy.fromfile(ff, dtype=pad, count=1)[0]['header']
            ff.seek(256 + 4, 0)
            hsize2 = numpy.fromfile(ff, dtype='i4', count=1)[0]
            if hsize2 != 256:
                raise IOError("trailing header marker mismatch")

            offsets = {}
            dtypes = {}
            ptr = 256 + 4 + 4
            for column, spec, ptypes in columndefs:
                if not isinstance(spec, tuple):
                    spec = (spec, ())
                shape = spec[1]
                if not isinstance(shape, tuple):
                    shape = (shape,) if shape else ()
                spec = (spec[0], shape)
                if ptypes == 'all':
                    ptypes = [0, 1, 2, 3, 4, 5]
                elif column == 'Mass':
                    ptypes = (header['Massarr'] == 0).nonzero()[0]

                reloffset = 0
                N = 0
                for i in ptypes:
                    if i == ptype:
                        reloffset = N
                    N += int(header['Npart'][i])

                prec = None
                if N != 0:
                    ff.seek(ptr, 0)
                    a = int(numpy.fromfile(ff, dtype='i4', count=1)[0])
                    ptr += 4
                    itemsize = a // N
                    offsets[column] = ptr + reloffset * itemsize
                    ptr += a
                    ff.seek(ptr, 0)
                    b = int(numpy.fromfile(ff, dtype='i4', count=1)[0])
                    ptr += 4
                    if a != b or b != N * itemsize:
                        raise IOError(
                            "F77 block markers for `%s` disagree: "
                            "start=%d end=%d truth=%d"
                            % (column, a, b, N * itemsize))
                    nmemb = int(numpy.prod(spec[1])) if spec[1] else 1
                    prec = itemsize // nmemb

                if spec[0] == 'auto':
                    if column == 'ID':
                        mapping = {8: 'i8', 4: 'i4', None: 'i4'}
                    else:
                        mapping = {8: 'f8', 4: 'f4', None: 'f4'}
                    spec = (mapping[prec], spec[1])

                if column == 'Mass' or ptype in ptypes:
                    dtypes[column] = (numpy.dtype(spec[0]), spec[1])
                    if column not in offsets:
                        offsets[column] = None   # broadcast Massarr

        size = int(header['Npart'][ptype])
        header_mass = float(header['Massarr'][ptype])

        from nbodykit_amd import CurrentMPIComm
        comm = comm if comm is not None else CurrentMPIComm.get()
        start, end = _partition(size, comm)
        self._size = end - start

        CatalogSource.__init__(self, comm=comm)
        for key in header.dtype.names:
            self.attrs[key] = numpy.array(header[key]).copy()
        if attrs is not None:
            self.attrs.update(attrs)

        with open(path, 'rb') as ff:
            for col, (base, shape) in dtypes.items():
                nmemb = int(numpy.prod(shape)) if shape else 1
                if offsets.get(col) is None:
                    data = numpy.full(end - start, header_mass,
                                      dtype=base)
                else:
                    ff.seek(offsets[col] + start * base.itemsize * nmemb)
                    data = numpy.fromfile(ff, dtype=base,
                                          count=(end - start) * nmemb)
                    if shape:
                        data = data.reshape((end - start,) + shape)
                self._overrides[col] = data
