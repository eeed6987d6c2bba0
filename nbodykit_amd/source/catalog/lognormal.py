"""
LogNormalCatalog (reference nbodykit/source/catalog/lognormal.py:9-191):
biased Poisson sample of a lognormal density field with Zel'dovich
displacements and velocities.
"""
import logging

import numpy

from nbodykit_amd import CurrentMPIComm
from nbodykit_amd import mockmaker
from nbodykit_amd.base.catalog import CatalogSource, column


class LogNormalCatalog(CatalogSource):
    logger = logging.getLogger("LogNormalCatalog")

    def __repr__(self):
        return ("LogNormalCatalog(seed=%(seed)d, cosmo_seed=%(cosmo_seed)d,"
                " bias=%(bias)g)" % self.attrs)

    @CurrentMPIComm.enable
    def __init__(self, Plin, nbar, BoxSize, Nmesh, bias=2., seed=None,
                 cosmo_seed=None, cosmo=None, redshift=None,
                 unitary_amplitude=False, inverted_phase=False, comm=None):
        self.comm = comm
        self.Plin = Plin

        if cosmo is None:
            cosmo = getattr(self.Plin, 'cosmo', None)
        if redshift is None:
            redshift = getattr(self.Plin, 'redshift', None)
        if cosmo is None:
            raise ValueError("'cosmo' must be passed if 'Plin' does not "
                             "have 'cosmo' attribute")
        if redshift is None:
            raise ValueError("'redshift' must be passed if 'Plin' does not "
                             "have 'redshift' attribute")
        self.cosmo = cosmo

        if hasattr(Plin, 'attrs'):
            self.attrs.update(Plin.attrs)
        else:
            self.attrs['cosmo'] = dict(cosmo.pars)

        self.attrs['nbar'] = nbar
        self.attrs['redshift'] = redshift
        self.attrs['bias'] = bias
        self.attrs['unitary_amplitude'] = unitary_amplitude
        self.attrs['inverted_phase'] = inverted_phase

        if seed is None:
            if self.comm.rank == 0:
                seed = numpy.random.randint(0, 4294967295)
                if cosmo_seed is None:
                    cosmo_seed = seed
            cosmo_seed = self.comm.bcast(cosmo_seed)
            seed = self.comm.bcast(seed)
        elif cosmo_seed is None:
            cosmo_seed = seed
        self.attrs['cosmo_seed'] = cosmo_seed
        self.attrs['seed'] = seed

        self._source = self._makesource(BoxSize=BoxSize, Nmesh=Nmesh)

        _Nmesh = numpy.empty(3, dtype='i8')
        _Nmesh[:] = Nmesh
        _Box = numpy.empty(3, dtype='f8')
        _Box[:] = BoxSize
        self.attrs['Nmesh'] = _Nmesh
        self.attrs['BoxSize'] = _Box

        self._size = len(self._source)
        CatalogSource.__init__(self, comm=comm)

        if self.csize == 0:
            raise ValueError("no particles in LogNormal source; try "
                             "increasing ``nbar`` parameter")

    @column
    def Position(self):
        """Position in Mpc/h"""
        return self.make_column(self._source['Position'])

    @column
    def Velocity(self):
        """Velocity in km/s"""
        return self.make_column(self._source['Velocity'])

    @column
    def VelocityOffset(self):
        """RSD offset f*psi, in Mpc/h"""
        return self.make_column(self._source['VelocityOffset'])

    def _makesource(self, BoxSize, Nmesh):
        _Nmesh = numpy.empty(3, dtype='i8')
        _Nmesh[:] = Nmesh
        _Box = numpy.empty(3, dtype='f8')
        _Box[:] = BoxSize

        # growth rate for the Zel'dovich velocities (lognormal.py:148)
        f = self.cosmo.scale_independent_growth_rate(self.attrs['redshift'])
        if self.comm.rank == 0:
            self.logger.info("Growth Rate is %g" % f)

        delta, disp = mockmaker.gaussian_real_fields(
            _Nmesh, _Box, self.Plin, int(self.attrs['cosmo_seed']),
            unitary_amplitude=self.attrs['unitary_amplitude'],
            inverted_phase=self.attrs['inverted_phase'],
            compute_displacement=True)

        pos, disp = mockmaker.poisson_sample_to_points(
            delta, disp, _Nmesh, _Box, self.attrs['nbar'], self.comm,
            bias=self.attrs['bias'], seed=self.attrs['seed'])

        # Zel'dovich move + velocity (lognormal.py:171-178)
        pos[:] = (pos + disp) % _Box
        z = self.attrs['redshift']
        velocity_norm = f * 100 * self.cosmo.efunc(z) / (1 + z)

        dtype = numpy.dtype([
            ('Position', ('f4', 3)),
            ('Velocity', ('f4', 3)),
            ('VelocityOffset', ('f4', 3)),
        ])
        source = numpy.empty(len(pos), dtype)
        source['Position'][:] = pos
        source['Velocity'][:] = velocity_norm * disp
        source['VelocityOffset'][:] = f * disp
        return source
