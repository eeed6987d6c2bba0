"""
Named mesh filters for ``MeshSource.apply`` (reference
nbodykit/filters.py:5-57).  These run through the generic host-evaluated
apply hook (ComplexField.apply, kind='wavenumber').
"""
import numpy

from nbodykit_amd.base.mesh import MeshFilter


class TopHat(MeshFilter):
    """Fourier-space top-hat of radius r (reference :5-33; ringing in
    configuration space is expected — it is the reference's behavior)."""
    kind = 'wavenumber'
    mode = 'complex'

    def __init__(self, r):
        self.r = r

    def filter(self, k, v):
        r = self.r
        k = sum(ki ** 2 for ki in k) ** 0.5
        kr = k * r
        with numpy.errstate(invalid='ignore', divide='ignore'):
            w = 3 * (numpy.sin(kr) / kr ** 3 - numpy.cos(kr) / kr ** 2)
        w[k == 0] = 1.0
        return w * v


class Gaussian(MeshFilter):
    """G(k) = exp(-0.5 k^2 r^2) (reference :35-57)."""
    kind = 'wavenumber'
    mode = 'complex'

    def __init__(self, r):
        self.r = r

    def filter(self, k, v):
        k2 = sum(ki ** 2 for ki in k)
        return numpy.exp(-0.5 * k2 * self.r ** 2) * v
