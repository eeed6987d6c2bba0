"""
MeshSlab / SlabIterator (reference nbodykit/meshtools.py:3-261) —
iteration over y-z planes of a coordinate mesh with norm2/mu and the
Hermitian double-count weights.  Part of the public API surface; the
product FFTPower does this reduction in the HIP binning kernel, but the
iterator is provided for user code that consumes ``field.x`` directly.
"""
import numpy


class MeshSlab(object):

    def __init__(self, islab, coords, axis, symmetry_axis):
        self.ndim = len(coords)
        self._index = islab
        self.axis = axis
        self.symmetry_axis = symmetry_axis
        self._coords = coords
        if self.hermitian_symmetric and self.symmetry_axis < 0:
            raise ValueError("`symmetry_axis` in MeshSlab must be "
                             "non-negative")

    def __str__(self):
        return "<%s: axis=%d, index=%d>" % (self.__class__.__name__,
                                            self.axis, self._index)

    __repr__ = __str__

    @property
    def index(self):
        toret = [slice(None)] * self.ndim
        toret[self.axis] = self._index
        return tuple(toret)

    @property
    def meshshape(self):
        return tuple(numpy.shape(self._coords[i])[i]
                     for i in range(self.ndim))

    @property
    def shape(self):
        return tuple(s for i, s in enumerate(self.meshshape)
                     if i != self.axis)

    @property
    def hermitian_symmetric(self):
        return self.symmetry_axis is not None

    def coords(self, i):
        if i < 0:
            i += self.ndim
        assert 0 <= i < self.ndim
        if i != self.axis:
            return numpy.take(self._coords[i], 0, axis=self.axis)
        return numpy.take(self._coords[i], self._index, axis=self.axis)

    def norm2(self):
        return sum(self.coords(i) ** 2 for i in range(self.ndim))

    def mu(self, los):
        norm = self.norm2() ** 0.5
        with numpy.errstate(invalid='ignore', divide='ignore'):
            result = sum(self.coords(i) * los[i]
                         for i in range(self.ndim)) / norm
        result[norm == 0.0] = 0.0
        return result

    @property
    def nonsingular(self):
        """True where the symmetry-axis frequency is strictly positive
        (DC and the negative-stored Nyquist excluded; reference :144-186)."""
        try:
            return self._nonsingular
        except AttributeError:
            if self.symmetry_axis == self.axis:
                if float(self.coords(self.axis)) <= 0.:
                    idx = numpy.zeros(self.shape, dtype=bool)
                else:
                    idx = numpy.ones(self.shape, dtype=bool)
            else:
                pos = self._coords[self.symmetry_axis] > 0.
                pos = numpy.take(pos, 0, axis=self.axis)
                idx = numpy.ones(self.shape, dtype=bool)
                idx[...] = pos
            self._nonsingular = idx
            return self._nonsingular

    @property
    def hermitian_weights(self):
        """2 on the doubled (positive-frequency) modes, else 1
        (reference :188-215)."""
        try:
            return self._weights
        except AttributeError:
            if not self.hermitian_symmetric:
                toret = 1.
            elif self.axis == self.symmetry_axis:
                toret = 2. if float(self.coords(self.symmetry_axis)) > 0. \
                    else 1.
            else:
                toret = numpy.ones(self.shape, dtype='f4')
                toret[self.nonsingular] = 2.
            self._weights = toret
            return self._weights


def SlabIterator(coords, axis=0, symmetry_axis=None):
    """Yield a MeshSlab per plane along ``axis`` (reference :217-261)."""
    ndim = len(coords)
    if ndim not in (2, 3):
        raise NotImplementedError("SlabIterator can only be used on 3D or "
                                  "2D arrays")
    if axis < 0:
        axis += ndim
    assert 0 <= axis < ndim
    if symmetry_axis is not None and symmetry_axis < 0:
        symmetry_axis += ndim

    shapes = [numpy.shape(x) for x in coords]
    try:
        mesh_size = [shape[i] for i, shape in enumerate(shapes)]
        for i in range(ndim):
            want = numpy.ones(ndim)
            want[i] = mesh_size[i]
            if shapes[i] != tuple(want):
                raise ValueError("coordinate array shape mismatch")
    except Exception:
        raise ValueError("input coordinates with shapes %s are not correct"
                         % str(shapes))

    N = numpy.shape(coords[axis])[axis]
    for islab in range(N):
        yield MeshSlab(islab, coords, axis, symmetry_axis)
