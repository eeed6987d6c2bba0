"""
CatalogSource — the particle-container side of the drop-in boundary
(reference nbodykit/base/catalog.py).  The reference's columns are lazy
dask arrays; dask is an implementation detail, not API (columns only
ever reach the mesh layer as numpy via ``source.compute(columns)``,
reference source/mesh/catalog.py:245), so here a :class:`ColumnAccessor`
is a numpy-backed column with the same access protocol:

- ``__getitem__`` dispatch order overrides > hardcolumns > defaults
  (reference :370-390)
- default columns ``Selection`` (True), ``Weight`` (1.0), ``Value`` (1.0)
  (reference :1166-1216)
- ``__setitem__`` with scalar -> ConstantArray broadcast (:955-972)
- boolean/slice selection returning a new catalog (:275-325)
- ``size`` / ``csize`` (:996-1011), ``compute`` (:530-560)
- ``to_mesh(...)`` (:787-873) with the same signature and defaults
"""
import logging

import numpy

from nbodykit_amd import CurrentMPIComm


def _is_torch(obj):
    return type(obj).__module__.startswith('torch')


def ConstantArray(value, size, chunks=None):
    """Zero-stride broadcast of a scalar to ``size`` rows (reference
    nbodykit/transform.py:89-107)."""
    value = numpy.asarray(value)
    return numpy.lib.stride_tricks.as_strided(
        value, (size,) + value.shape, (0,) + value.strides)


class ColumnAccessor(numpy.ndarray):
    """A numpy view carrying its catalog and attrs (the reference's is a
    dask.Array subclass, nbodykit/base/catalog.py:12-95)."""

    def __new__(cls, catalog, array, is_default=False):
        obj = numpy.asarray(array).view(cls)
        obj.catalog = catalog
        obj.is_default = is_default
        obj.attrs = {}
        return obj

    def __array_finalize__(self, obj):
        if obj is None:
            return
        self.catalog = getattr(obj, 'catalog', None)
        self.is_default = getattr(obj, 'is_default', False)
        self.attrs = dict(getattr(obj, 'attrs', {}))

    def compute(self):
        return numpy.asarray(self)

    def as_numpy(self):
        return numpy.asarray(self)

    # in-place operators DISOWN instead of mutating: the reference's
    # dask-backed columns never write back to the catalog (an operation
    # yields a new array; ``cat[col] *= 10`` works via __setitem__) —
    # a live numpy view would silently corrupt the stored column.
    def __iadd__(self, other):
        return numpy.asarray(self) + other

    def __isub__(self, other):
        return numpy.asarray(self) - other

    def __imul__(self, other):
        return numpy.asarray(self) * other

    def __itruediv__(self, other):
        return numpy.asarray(self) / other

    def __ifloordiv__(self, other):
        return numpy.asarray(self) // other

    def __ipow__(self, other):
        return numpy.asarray(self) ** other

    def __imod__(self, other):
        return numpy.asarray(self) % other


def column(name=None, is_default=False):
    """Decorator marking a method as a named column (reference :97-125)."""
    def decorator(getter, name=name):
        getter.column_name = getter.__name__ if name is None else name
        getter.is_default = is_default
        return getter
    if callable(name):
        getter = name
        return decorator(getter, name=getter.__name__)
    return decorator


def find_columns(cls):
    """Collect decorated columns from the class hierarchy (the reference
    does this with a metaclass, ColumnFinder, :127-160)."""
    hard, defaults = set(), set()
    for klass in cls.__mro__:
        for value in vars(klass).values():
            if callable(value) and hasattr(value, 'column_name'):
                (defaults if value.is_default else hard).add(
                    value.column_name)
    return hard, defaults


class CatalogSourceBase(object):
    logger = logging.getLogger('CatalogSourceBase')

    def __new__(cls, *args, **kwargs):
        obj = object.__new__(cls)
        obj._overrides = {}
        obj.base = None
        hard, defaults = find_columns(cls)
        obj._hardcolumns = sorted(hard)
        obj._defaultcolumns = sorted(defaults)
        return obj

    def __init__(self, comm=None):
        self.comm = comm if comm is not None else CurrentMPIComm.get()

    @property
    def attrs(self):
        try:
            return self._attrs
        except AttributeError:
            self._attrs = {}
            return self._attrs

    # -- column protocol --------------------------------------------------
    @property
    def hardcolumns(self):
        return self._hardcolumns

    @property
    def columns(self):
        all_cols = set(self._overrides) | set(self._hardcolumns) \
            | set(self._defaultcolumns)
        return sorted(all_cols)

    def __contains__(self, col):
        return col in self.columns

    def __getitem__(self, sel):
        if isinstance(sel, str):
            # dispatch order: overrides > hardcolumns > defaults (:370-390)
            if sel in self._overrides:
                arr = self._overrides[sel]
                is_default = False
            elif sel in self._hardcolumns:
                arr = getattr(self, sel)()
                is_default = False
            elif sel in self._defaultcolumns:
                arr = getattr(self, sel)()
                is_default = True
            else:
                raise KeyError("column `%s` is not defined in this source; "
                               "try adding column via `source[column] = data`"
                               % sel)
            if isinstance(arr, ColumnAccessor) or _is_torch(arr):
                # device-resident columns pass through untouched
                return arr
            return ColumnAccessor(self, arr, is_default=is_default)

        # boolean mask or slice -> a new catalog view (:275-325)
        if isinstance(sel, slice) or (
                isinstance(sel, (numpy.ndarray, list))):
            return self._get_slice(sel)
        raise KeyError("unsupported selection type: %r" % (sel,))

    def __setitem__(self, col, value):
        size = getattr(self, 'size', None)
        if numpy.isscalar(value):
            assert size is not None, "size must be known to set a scalar"
            value = ConstantArray(value, size)
        if not _is_torch(value):
            value = numpy.asarray(value)
        if size is not None and len(value) != size:
            raise ValueError(
                "error setting column '%s': data of shape %s does not match "
                "catalog size %d" % (col, value.shape, size))
        self._overrides[col] = value

    def __delitem__(self, col):
        if col not in self.columns:
            raise ValueError("no such column, cannot delete it")
        if col in self._overrides:
            del self._overrides[col]
        elif col in self.hardcolumns:
            raise ValueError("cannot delete a hard-coded column")
        else:
            # a default column (Selection/Weight/Value) with no override
            # is not deletable (reference base/catalog.py:944-953 raises
            # for non-overridable columns; a silent no-op hides typos)
            raise ValueError(
                "cannot delete default column '%s' (no override set)" % col)

    def _get_slice(self, index):
        if isinstance(index, list):
            index = numpy.asarray(index)
        if isinstance(index, numpy.ndarray) and index.dtype == bool:
            if len(index) != self.size:
                raise ValueError("boolean mask must have catalog length")
        subset = object.__new__(CatalogSource)
        subset._overrides = {}
        subset.base = None
        subset._hardcolumns = []
        subset._defaultcolumns = list(self._defaultcolumns)
        subset.comm = self.comm
        subset._attrs = dict(self.attrs)
        # materialize every non-default column, sliced
        n = None
        for col in self.columns:
            if col in self._defaultcolumns and col not in self._overrides:
                continue
            data = numpy.asarray(self[col])[index]
            subset._overrides[col] = data
            n = len(data)
        if n is None:
            n = len(numpy.zeros(self.size)[index])
        subset._size = n
        subset._csize = subset.comm.allreduce(n)
        return subset

    def compute(self, *args):
        """Materialize columns to numpy (reference :530-560; trivial for
        numpy-backed columns but kept for API parity).  A single list
        argument is materialized element-wise, like dask.compute."""
        def materialize(a):
            return a if _is_torch(a) else numpy.asarray(a)
        if len(args) == 1 and isinstance(args[0], (list, tuple)):
            return [materialize(a) for a in args[0]]
        toret = tuple(materialize(a) for a in args)
        if len(toret) == 1:
            return toret[0]
        return list(toret)

    def make_column(self, array):
        return numpy.asarray(array)

    def view(self, type=None):
        """A view sharing columns and attrs with ``self`` (write-through
        — adding a column on the view adds it to the source; reference
        :440-472).  ``type`` optionally rebrands the class."""
        cls = self.__class__ if type is None else type
        obj = CatalogSourceBase.__new__(cls)
        obj.__dict__.update(self.__dict__)
        obj._overrides = self._overrides      # shared, not copied
        obj.base = self
        return obj

    def copy(self):
        """Shallow copy: every column of ``self`` referenced (no data
        copied), attrs decoupled (reference :474-507).  The copy is a
        plain CatalogSource holding the materialized column references
        (the reference keeps the subclass; its hard columns are lazy
        dask graphs — ours are arrays either way)."""
        toret = object.__new__(CatalogSource)
        toret._overrides = {}
        toret.base = None
        toret._hardcolumns = []
        toret._defaultcolumns = list(self._defaultcolumns)
        toret.comm = self.comm
        toret._size = self.size
        toret._csize = self.csize
        for col in self.columns:
            if col in self._defaultcolumns and col not in self._overrides:
                continue
            toret._overrides[col] = self[col]
        toret._attrs = dict(self.attrs)
        return toret

    def gslice(self, start, stop, end=1, redistribute=True):
        """Global slice across ranks (reference :1013-1077); with
        ``redistribute`` the selected rows are re-balanced evenly."""
        counts = self.comm.allgather(self.size)
        offset = int(numpy.sum(counts[:self.comm.rank], dtype='i8'))
        index = numpy.zeros(self.csize, dtype=bool)
        index[slice(start, stop, end)] = True
        sub = self[index[offset:offset + self.size]]
        if redistribute and self.comm.size > 1:
            rank, ws = self.comm.rank, self.comm.size
            n_tot = sub.csize
            lo = n_tot * rank // ws
            hi = n_tot * (rank + 1) // ws
            data = {}
            for col in sub.columns:
                if col in sub._defaultcolumns                         and col not in sub._overrides:
                    continue
                full = numpy.concatenate(
                    self.comm.allgather(numpy.asarray(sub[col])), axis=0)
                data[col] = full[lo:hi]
            from nbodykit_amd.source.catalog.array import ArrayCatalog
            out = ArrayCatalog(data, comm=self.comm) if data else sub
            out.attrs.update(sub.attrs)
            return out
        return sub

    def save(self, output, columns=None, dataset=None, datasets=None,
             header='Header', compute=True):
        """Save the catalog to a bigfile directory (reference
        :562-695): one block per non-default column (f8 etc. as stored,
        32Mi rows per physical file, rank rows ordered by rank), column
        attrs on the blocks, and :attr:`attrs` on the ``header`` block
        with the ``json://`` fallback for non-array values."""
        import json
        from nbodykit_amd.io.bigfile_format import BigFile
        from nbodykit_amd.utils import JSONEncoder

        if columns is None:
            columns = self.columns
        columns = [col for col in columns if not self[col].is_default]

        if datasets is not None:
            import warnings
            warnings.warn("datasets argument is deprecated. Specify a "
                          "single dataset prefix for all columns instead.")
        elif dataset is not None:
            datasets = [dataset + '/' + col for col in columns]
        else:
            datasets = list(columns)
        if len(datasets) != len(columns):
            raise ValueError("`datasets` must have the same length as "
                             "`columns`")

        with BigFile(output, create=True, comm=self.comm) as ff:
            for column, ds in zip(columns, datasets):
                array = self[column]
                if _is_torch(array):
                    array = array.cpu().numpy()
                array = numpy.asarray(array)
                bb = ff.create_from_array(ds, array)
                if hasattr(self[column], 'attrs'):
                    for key, v in self[column].attrs.items():
                        bb.attrs[key] = v

            if header is not None:
                bb = ff.create(header)
                for key in self.attrs:
                    value = self.attrs[key]
                    try:
                        bb.attrs[key] = value
                    except (ValueError, TypeError):
                        try:
                            bb.attrs[key] = 'json://' + json.dumps(
                                value, cls=JSONEncoder)
                        except Exception:
                            raise ValueError(
                                "cannot save '%s' key in attrs "
                                "dictionary" % key)

    @property
    def Index(self):
        """The global row index, as an ATTRIBUTE like the reference's
        (base/catalog.py:1178-1195 exposes a property, not a column):
        offset of the lower ranks + local arange, dtype i8."""
        counts = self.comm.allgather(self.size)
        offset = int(numpy.sum(counts[:self.comm.rank], dtype='i8'))
        return offset + numpy.arange(self.size, dtype='i8')

    # -- default columns (reference :1166-1216) ---------------------------
    @column(is_default=True)
    def Selection(self):
        return ConstantArray(True, self.size)

    @column(is_default=True)
    def Weight(self):
        return ConstantArray(1.0, self.size)

    @column(is_default=True)
    def Value(self):
        return ConstantArray(1.0, self.size)

    # -- mesh view --------------------------------------------------------
    def to_mesh(self, Nmesh=None, BoxSize=None, dtype='f4', interlaced=False,
                compensated=False, resampler='cic', weight='Weight',
                value='Value', selection='Selection', position='Position',
                window=None):
        """Create a CatalogMesh view (reference :787-873; same defaults:
        dtype='f4', resampler='cic', compensated=False, interlaced=False)."""
        from nbodykit_amd.source.mesh.catalog import CatalogMesh

        if window is not None:
            import warnings
            warnings.warn("The window argument is deprecated. Use "
                          "`resampler=` instead", DeprecationWarning,
                          stacklevel=2)
            resampler = window

        for col in [weight, selection]:
            if col not in self:
                raise ValueError("column '%s' missing; cannot create mesh"
                                 % col)
        if resampler not in ('cic', 'tsc', 'pcs'):
            raise ValueError("valid resampler: ['cic', 'tsc', 'pcs']")

        if BoxSize is None:
            try:
                BoxSize = self.attrs['BoxSize']
            except KeyError:
                raise ValueError(
                    "cannot convert particle source to a mesh; 'BoxSize' "
                    "keyword is not supplied and the CatalogSource does not "
                    "define one in 'attrs'.")
        if Nmesh is None:
            try:
                Nmesh = self.attrs['Nmesh']
            except KeyError:
                raise ValueError(
                    "cannot convert particle source to a mesh; 'Nmesh' "
                    "keyword is not supplied and the CatalogSource does not "
                    "define one in 'attrs'.")

        return CatalogMesh(self, Nmesh=Nmesh, BoxSize=BoxSize, dtype=dtype,
                           Weight=self[weight], Selection=self[selection],
                           Value=self[value], Position=self[position],
                           interlaced=interlaced, compensated=compensated,
                           resampler=resampler)


class CatalogSource(CatalogSourceBase):
    """A catalog with a well-defined local ``size`` (reference :875-1011)."""

    def __init__(self, comm=None):
        CatalogSourceBase.__init__(self, comm=comm)
        if not hasattr(self, '_size'):
            raise ValueError("CatalogSource subclasses must set _size "
                             "before calling __init__")
        self._csize = self.comm.allreduce(self._size)

    @property
    def size(self):
        return self._size

    @property
    def csize(self):
        return self._csize

    def __len__(self):
        return self.size

    def __repr__(self):
        return "%s(size=%d)" % (type(self).__name__, self.size)
