"""
BinnedStatistic — the result container (reference
nbodykit/binned_statistic.py:60-955): an xarray-like grid of binned
variables with named dims, edges, coordinates, NaN masking, selection
(sel/take/squeeze), re-binning (reindex/average) and the JSON round-trip
through nbodykit's ``__dtype__/__shape__/__data__`` schema
(utils.py:381-489) — files interchange with real nbodykit.
"""
import numpy


def bin_ndarray(ndarray, new_shape, weights=None, operation=numpy.mean):
    """Re-bin an array to an integer-factor smaller shape by applying
    ``operation`` over the collapsed sub-blocks (reference :3-58)."""
    if ndarray.shape == new_shape:
        raise ValueError("why are we re-binning if the new shape equals "
                         "the old shape?")
    if ndarray.ndim != len(new_shape):
        raise ValueError("Shape mismatch: {} -> {}".format(
            ndarray.shape, new_shape))
    if numpy.any(numpy.mod(ndarray.shape, new_shape)):
        raise ValueError("desired shape of %s must be integer factor "
                         "smaller than the old shape %s"
                         % (str(new_shape), str(ndarray.shape)))

    pairs = [(d, c // d) for d, c in zip(new_shape, ndarray.shape)]
    flattened = [l for p in pairs for l in p]
    ndarray = ndarray.reshape(flattened)
    if weights is not None:
        weights = weights.reshape(flattened)

    for i in range(len(new_shape)):
        if weights is not None:
            ndarray = operation(ndarray * weights, axis=-1 * (i + 1))
            weights = numpy.sum(weights, axis=-1 * (i + 1))
            ndarray /= weights
        else:
            ndarray = operation(ndarray, axis=-1 * (i + 1))
    return ndarray


class BinnedStatistic(object):

    def __init__(self, dims, edges, data, fields_to_sum=[], coords=None,
                 **kwargs):
        if len(dims) != len(edges):
            raise ValueError("size mismatch between specified `dims` and "
                             "`edges`")
        if not isinstance(data, numpy.ndarray) or data.dtype.names is None:
            raise TypeError("'data' should be a structured numpy array")

        shape = tuple(len(e) - 1 for e in edges)
        if data.shape != shape:
            raise ValueError("`edges` imply data shape of %s, but data has "
                             "shape %s" % (shape, data.shape))

        self.dims = list(dims)
        self.edges = dict(zip(self.dims, edges))

        self.coords = {}
        for i, dim in enumerate(self.dims):
            if coords is not None and coords[i] is not None:
                self.coords[dim] = numpy.copy(coords[i])
            else:
                self.coords[dim] = 0.5 * (edges[i][1:] + edges[i][:-1])

        self.data = data.copy()

        # mask bins where any variable is non-finite
        self.mask = numpy.zeros(self.shape, dtype=bool)
        for name in data.dtype.names:
            self.mask |= ~numpy.isfinite(self.data[name])

        self._fields_to_sum = list(fields_to_sum)

        self.attrs = {}
        for k in kwargs:
            self.attrs[k] = kwargs[k]

    # -- state / io -------------------------------------------------------
    @classmethod
    def from_state(cls, state):
        obj = cls(dims=state['dims'], edges=state['edges'],
                  coords=state['coords'], data=state['data'])
        obj.attrs.update(state['attrs'])
        return obj

    def __getstate__(self):
        return dict(dims=self.dims,
                    edges=[self.edges[d] for d in self.dims],
                    coords=[self.coords[d] for d in self.dims],
                    data=self.data,
                    attrs=self.attrs)

    def to_json(self, filename):
        import json
        from nbodykit_amd.utils import JSONEncoder
        with open(filename, 'w') as ff:
            json.dump(self.__getstate__(), ff, cls=JSONEncoder)

    @classmethod
    def from_json(cls, filename, key='data', dims=None, edges=None,
                  **kwargs):
        import json
        from nbodykit_amd.utils import JSONDecoder
        with open(filename, 'r') as ff:
            state = json.load(ff, cls=JSONDecoder)

        if key not in state:
            raise ValueError("no data entry found in JSON format for '%s' "
                             "key; valid keys are %s"
                             % (key, tuple(state.keys())))
        data = state[key]
        dims = state.pop('dims', dims)
        if dims is None:
            raise ValueError("no `dims` found in JSON file; please specify "
                             "as keyword argument")
        edges = state.pop('edges', edges)
        if edges is None:
            raise ValueError("no `edges` found in JSON file; please "
                             "specify as keyword argument")
        coords = state.pop('coords', None)
        attrs = state.pop('attrs', {})
        attrs.update(kwargs)
        return cls(dims, edges, data, coords=coords, **attrs)

    # -- basic protocol ---------------------------------------------------
    @property
    def shape(self):
        return tuple(len(self.coords[d]) for d in self.dims)

    @property
    def variables(self):
        return list(self.data.dtype.names)

    def __str__(self):
        name = self.__class__.__name__
        dims = "(" + ", ".join('%s: %d' % (k, self.shape[i])
                               for i, k in enumerate(self.dims)) + ")"
        if len(self.variables) < 5:
            return "<%s: dims: %s, variables: %s>" \
                % (name, dims, str(tuple(self.variables)))
        return "<%s: dims: %s, variables: %d total>" \
            % (name, dims, len(self.variables))

    __repr__ = __str__

    def __iter__(self):
        return iter(self.variables)

    def __contains__(self, key):
        return key in self.variables

    def __setitem__(self, key, value):
        if numpy.shape(value) != self.data.shape:
            raise ValueError("data to be added must have shape %s"
                             % str(self.data.shape))
        dtype = list(self.data.dtype.descr)
        names = list(self.data.dtype.names)
        if key in names:
            i = names.index(key)
            dtype.pop(i)
            names.pop(i)
        dtype += [(key, numpy.asarray(value).dtype.type)]
        new = numpy.zeros(self.data.shape, dtype=numpy.dtype(dtype))
        for col in names:
            new[col] = self.data[col]
        new[key] = value
        self.data = new
        self.mask = self.mask | ~numpy.isfinite(new[key])

    def __getitem__(self, key):
        if isinstance(key, str):
            if key in self.variables:
                return self.data[key]
            raise KeyError("`%s` is not a valid variable name" % key)

        indices = [list(range(self.shape[i]))
                   for i in range(len(self.dims))]

        if isinstance(key, (list, tuple)) \
                and all(isinstance(x, str) for x in key):
            if all(k in self.variables for k in key):
                return self.__finalize__(self.data[list(key)],
                                         self.mask.copy(), indices)
            bad = ', '.join("'%s'" % k for k in key
                            if k not in self.variables)
            raise KeyError("cannot slice variables -- invalid names: (%s)"
                           % bad)

        key_ = key
        if isinstance(key, (slice, int)) or (
                isinstance(key, list)
                and all(isinstance(x, int) for x in key)):
            key_ = [key]

        squeezed = []
        for i, subkey in enumerate(key_):
            if i >= len(self.dims):
                raise IndexError("too many indices for BinnedStatistic; "
                                 "note that ndim = %d" % len(self.dims))
            if isinstance(subkey, int):
                indices[i] = [subkey]
                squeezed.append(self.dims[i])
            elif isinstance(subkey, list):
                indices[i] = subkey
            elif isinstance(subkey, slice):
                indices[i] = list(range(*subkey.indices(self.shape[i])))

        if len(squeezed) == len(self.dims):
            raise IndexError("cannot return object with all remaining "
                             "dimensions squeezed")

        try:
            toret = self.__finalize__(self.data[key], self.mask[key],
                                      indices)
            for dim in squeezed:
                toret = toret.squeeze(dim)
            return toret
        except ValueError:
            raise IndexError("this type of slicing not implemented")

    # -- construction helpers --------------------------------------------
    @classmethod
    def __construct_direct__(cls, data, mask, **kwargs):
        obj = object.__new__(cls)
        for k in kwargs:
            setattr(obj, k, kwargs[k])
        for k, d in zip(['data', 'mask'], [data, mask]):
            setattr(obj, k, d)
            if obj.shape != d.shape:
                try:
                    setattr(obj, k, d.reshape(obj.shape))
                except Exception:
                    raise ValueError("shape mismatch between data and "
                                     "coordinates")
        return obj

    def __copy_attrs__(self):
        return dict(dims=list(self.dims), edges=self.edges.copy(),
                    coords=self.coords.copy(), attrs=self.attrs.copy(),
                    _fields_to_sum=list(self._fields_to_sum))

    def __finalize__(self, data, mask, indices):
        edges, coords = self.__slice_edges__(indices)
        kw = dict(dims=list(self.dims), edges=edges, coords=coords,
                  attrs=self.attrs.copy(),
                  _fields_to_sum=self._fields_to_sum)
        return self.__class__.__construct_direct__(data, mask, **kw)

    def __slice_edges__(self, indices):
        edges = {}
        coords = {}
        for i, dim in enumerate(self.dims):
            if len(indices[i]) > 0:
                idx = list(indices[i]) + [indices[i][-1] + 1]
            else:
                idx = [0]
            edges[dim] = self.edges[dim][idx]
            coords[dim] = 0.5 * (edges[dim][1:] + edges[dim][:-1])
        return edges, coords

    def copy(self, cls=None):
        attrs = self.__copy_attrs__()
        if cls is None:
            cls = self.__class__
        if not issubclass(cls, BinnedStatistic):
            raise TypeError("cls must be a BinnedStatistic subclass")
        return cls.__construct_direct__(self.data.copy(), self.mask.copy(),
                                        **attrs)

    def rename_variable(self, old_name, new_name):
        import copy as _copy
        if old_name not in self.variables:
            raise ValueError("`%s` is not an existing variable name"
                             % old_name)
        new_dtype = _copy.deepcopy(self.data.dtype)
        names = list(new_dtype.names)
        names[names.index(old_name)] = new_name
        new_dtype.names = names
        self.data.dtype = new_dtype

    # -- selection --------------------------------------------------------
    def _get_index(self, dim, val, method=None):
        index = self.coords[dim]
        if method == 'nearest':
            return int(numpy.abs(index - val).argmin())
        try:
            return list(index).index(val)
        except Exception as e:
            raise IndexError("error converting '%s' index; try setting "
                             "`method = 'nearest'`: %s" % (dim, str(e)))

    def sel(self, method=None, **indexers):
        indices = {}
        squeezed = []
        for dim, key in indexers.items():
            if isinstance(key, list):
                indices[dim] = [self._get_index(dim, k, method=method)
                                for k in key]
            elif isinstance(key, slice):
                lo = self._get_index(dim, key.start, method=method)
                hi = self._get_index(dim, key.stop, method=method)
                i = self.dims.index(dim)
                indices[dim] = list(range(*slice(lo, hi)
                                          .indices(self.shape[i])))
            elif not numpy.isscalar(key):
                raise IndexError("please index using a list, slice, or "
                                 "scalar value")
            else:
                indices[dim] = [self._get_index(dim, key, method=method)]
                squeezed.append(dim)

        if len(squeezed) == len(self.dims):
            raise IndexError("cannot return object with all remaining "
                             "dimensions squeezed")
        toret = self.take(**indices)
        for dim in squeezed:
            toret = toret.squeeze(dim)
        return toret

    def take(self, *masks, **indices_dict):
        mask = numpy.ones(self.shape, dtype='?')
        for m in masks:
            mask = mask & m

        indices = [numpy.ones(self.shape[i], dtype='?')
                   for i in range(len(self.dims))]
        for i, dim in enumerate(self.dims):
            axis = tuple(j for j in range(len(self.dims)) if j != i)
            indices[i] &= mask.all(axis=axis)

        for dim, index in indices_dict.items():
            i = self.dims.index(dim)
            index = numpy.asarray(index) if not numpy.isscalar(index) \
                else index
            if isinstance(index, numpy.ndarray) and index.dtype == bool:
                assert index.ndim == 1
                indices[i] &= index
            else:
                m1 = numpy.zeros(self.shape[i], dtype='?')
                m1.put(index, True)
                indices[i] &= m1

        indices = [idx.nonzero()[0] for idx in indices]

        data = self.data.copy()
        mask = self.mask.copy()
        for i, idx in enumerate(indices):
            data = numpy.take(data, idx, axis=i)
            mask = numpy.take(mask, idx, axis=i)
        return self.__finalize__(data, mask, indices)

    def squeeze(self, dim=None):
        if dim is None:
            dim = [k for k in self.dims if len(self.coords[k]) == 1]
            if not len(dim):
                raise ValueError("no available dimensions with length one "
                                 "to squeeze")
            if len(dim) > 1:
                raise ValueError("multiple dimensions available to squeeze "
                                 "-- please specify")
            dim = dim[0]
        else:
            if dim not in self.dims:
                raise ValueError("`%s` is not a valid dimension name" % dim)
            if len(self.coords[dim]) != 1:
                raise ValueError("the `%s` dimension must have length one "
                                 "to squeeze" % dim)

        i = self.dims.index(dim)
        toret = self.copy()
        toret.dims.pop(i)
        toret.edges.pop(dim)
        toret.coords.pop(dim)
        if not len(toret.dims):
            raise ValueError("cannot squeeze the only remaining axis")
        attrs = toret.__copy_attrs__()
        return self.__construct_direct__(toret.data.squeeze(axis=i),
                                         toret.mask.squeeze(axis=i),
                                         **attrs)

    # -- re-binning -------------------------------------------------------
    def average(self, dim, **kwargs):
        spacing = self.edges[dim][-1] - self.edges[dim][0]
        toret = self.reindex(dim, spacing, **kwargs)
        return toret.sel(**{dim: toret.coords[dim][0]})

    def reindex(self, dim, spacing, weights=None, force=True,
                return_spacing=False, fields_to_sum=[]):
        i = self.dims.index(dim)
        fields_to_sum = list(fields_to_sum) + self._fields_to_sum

        old_spacings = numpy.diff(self.coords[dim])
        old_spacing = old_spacings[0]

        factor = int(numpy.round(spacing / old_spacing))
        if not factor:
            raise ValueError("new spacing must be smaller than original "
                             "spacing of %.2e" % old_spacing)
        if factor == 1:
            raise ValueError("closest binning size to input spacing is the "
                             "same as current binning")
        if not numpy.allclose(old_spacing * factor, spacing) and not force:
            raise ValueError("if `force = False`, new bin spacing must be "
                             "an integral factor smaller than original")

        data = self.data.copy()
        if isinstance(weights, str):
            if weights not in self.variables:
                raise ValueError("cannot weight by `%s`; no such column"
                                 % weights)
            weights = self.data[weights]

        edges = self.edges[dim]
        new_shape = list(self.shape)

        leftover = self.shape[i] % factor
        if leftover and not force:
            raise ValueError("cannot re-bin because there are %d extra "
                             "bins, using spacing = %.2e"
                             % (leftover, old_spacing * factor))
        if leftover:
            sl = [slice(None)] * len(self.dims)
            sl[i] = slice(None, -leftover)
            data = data[tuple(sl)]
            if weights is not None:
                weights = weights[tuple(sl)]
            edges = edges[:-leftover]
            new_shape[i] -= leftover

        new_shape[i] = int(new_shape[i] // factor)
        new_edges = numpy.linspace(edges[0], edges[-1], new_shape[i] + 1)

        new_data = numpy.empty(tuple(new_shape), dtype=self.data.dtype)
        for name in self.variables:
            operation = numpy.nanmean
            w = weights
            if weights is not None or name in fields_to_sum:
                operation = numpy.nansum
                if name in fields_to_sum:
                    w = None
            new_data[name] = bin_ndarray(data[name], tuple(new_shape),
                                         weights=w, operation=operation)

        new_mask = numpy.zeros(tuple(new_shape), dtype=bool)
        for name in self.variables:
            new_mask |= ~numpy.isfinite(new_data[name])

        kw = self.__copy_attrs__()
        kw['edges'][dim] = new_edges
        kw['coords'][dim] = 0.5 * (new_edges[1:] + new_edges[:-1])
        toret = self.__construct_direct__(new_data, new_mask, **kw)
        return (toret, spacing) if return_spacing else toret
