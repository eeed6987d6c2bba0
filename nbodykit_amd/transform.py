"""
Column transform helpers (reference nbodykit/transform.py) — the subset
the hot path and its tests touch: ConstantArray (:89-107, re-exported
from base.catalog) and the RSD helpers VectorProjection (:489-515) /
CartesianToEquatorial-style utilities are out of scope.
"""
import numpy

from nbodykit_amd.base.catalog import ConstantArray  # noqa: F401


def StackColumns(*cols):
    """Stack 1D columns into a (N, ncol) array (reference :30-63)."""
    cols = [numpy.asarray(c) for c in cols]
    return numpy.stack(cols, axis=-1)


def ConcatenateSources(*sources, **kwargs):
    raise NotImplementedError(
        "ConcatenateSources is outside the FFTPower hot-path scope")


def VectorProjection(vector, direction):
    """Components of ``vector`` along ``direction``:
    (v . d_hat) d_hat (reference :489-515)."""
    direction = numpy.asarray(direction, dtype='f8')
    direction = direction / (direction ** 2).sum() ** 0.5
    vector = numpy.asarray(vector)
    projection = (vector * direction).sum(axis=-1)
    return projection[:, None] * direction[None, :]
