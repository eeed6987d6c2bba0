"""
RedshiftHistogram — n(z) from a catalog of objects (reference
nbodykit/algorithms/zhist.py:9-252): weighted redshift histogram
normalized by comoving shell volumes (units (Mpc/h)^-3), with Scott's
rule for automatic binning and spline interpolation.  Host numpy —
this feeds the FKP ``NZ`` column, not the GPU path.
"""
import logging

import numpy

from nbodykit_amd import CurrentMPIComm
from nbodykit_amd.base.catalog import ConstantArray


class RedshiftHistogram(object):
    logger = logging.getLogger('RedshiftHistogram')

    def __init__(self, source, fsky, cosmo, bins=None,
                 redshift='Redshift', weight=None):
        for col in [redshift, weight]:
            if col is not None and col not in source:
                raise ValueError("'%s' column missing from input source "
                                 "in RedshiftHistogram" % col)
        self.comm = source.comm

        if bins is None:
            h, bins = scotts_bin_width(
                numpy.asarray(source[redshift]), self.comm)
            if self.comm.rank == 0:
                self.logger.info("using Scott's rule to determine "
                                 "optimal binning; h = %.2e, N_bins = %d"
                                 % (h, len(bins) - 1))
        elif numpy.isscalar(bins):
            z = numpy.asarray(source[redshift])
            maxval = self.comm.allreduce(z.max(), op='max')
            minval = self.comm.allreduce(z.min(), op='min')
            bins = numpy.linspace(minval, maxval, bins + 1, endpoint=True)

        self.source = source
        self.cosmo = cosmo

        self.attrs = {}
        self.attrs['edges'] = bins
        self.attrs['fsky'] = fsky
        self.attrs['redshift'] = redshift
        self.attrs['weight'] = weight

        self.run()

    def run(self):
        """Histogram + shell-volume normalization (reference :82-139);
        sets ``bin_edges``, ``bin_centers``, ``dV`` ((Mpc/h)^3) and
        ``nbar`` ((Mpc/h)^-3) on every rank."""
        edges = numpy.asarray(self.attrs['edges'], dtype='f8')

        redshift = numpy.asarray(self.source[self.attrs['redshift']])
        if self.attrs['weight'] is not None:
            weight = numpy.asarray(self.source[self.attrs['weight']])
        else:
            weight = ConstantArray(1.0, self.source.size)

        dig = numpy.searchsorted(edges, redshift, "right")
        N = numpy.bincount(dig, weights=weight,
                           minlength=len(edges) + 1)[1:-1]
        N = self.comm.allreduce(N)

        R_hi = self.cosmo.comoving_distance(edges[1:])
        R_lo = self.cosmo.comoving_distance(edges[:-1])
        dV = (4. / 3) * numpy.pi * (R_hi ** 3 - R_lo ** 3) \
            * self.attrs['fsky']

        self.bin_edges = edges
        self.bin_centers = 0.5 * (edges[:-1] + edges[1:])
        self.dV = dV
        self.nbar = 1. * N / dV

    def interpolate(self, z, ext='zeros'):
        """Spline n(z) at the given redshifts (reference :141-159) —
        the usual way to fill an FKP ``NZ`` column."""
        from scipy.interpolate import InterpolatedUnivariateSpline
        nofz = InterpolatedUnivariateSpline(self.bin_centers, self.nbar,
                                            ext=ext)
        return nofz(z)

    def __getstate__(self):
        return dict(bin_edges=self.bin_edges,
                    bin_centers=self.bin_centers,
                    dV=self.dV, nbar=self.nbar, attrs=self.attrs)

    def __setstate__(self, state):
        self.__dict__.update(state)

    def save(self, output):
        import json
        from nbodykit_amd.utils import JSONEncoder
        if self.comm.rank == 0:
            with open(output, 'w') as ff:
                json.dump(self.__getstate__(), ff, cls=JSONEncoder)

    @classmethod
    @CurrentMPIComm.enable
    def load(cls, output, comm=None):
        import json
        from nbodykit_amd.utils import JSONDecoder
        if comm.rank == 0:
            with open(output, 'r') as ff:
                state = json.load(ff, cls=JSONDecoder)
        else:
            state = None
        state = comm.bcast(state)
        self = object.__new__(cls)
        self.__setstate__(state)
        self.comm = comm
        return self


def scotts_bin_width(data, comm):
    """Optimal histogram bin width, h = sigma (24 sqrt(pi)/n)^(1/3)
    (reference :210-252; collective)."""
    csum = comm.allreduce(data.sum())
    csize = comm.allreduce(data.size)
    cmean = csum / csize
    rsum = comm.allreduce((abs(data - cmean) ** 2).sum())
    sigma = (rsum / csize) ** 0.5

    dx = sigma * (24. * numpy.sqrt(numpy.pi) / csize) ** (1. / 3)
    maxval = comm.allreduce(data.max(), op='max')
    minval = comm.allreduce(data.min(), op='min')

    Nbins = numpy.ceil((maxval - minval) * 1. / dx)
    Nbins = max(1, int(Nbins))
    edges = minval + dx * numpy.arange(Nbins + 1)
    return dx, edges
