"""
FKPCatalog — the data+randoms pair behind ConvolvedFFTPower (reference
nbodykit/algorithms/convpower/catalog.py:7-259; Feldman, Kaiser &
Peacock 1994): uniform access to both species, the shared Cartesian
bounding box from the randoms (+BoxPad), and ``to_mesh`` producing the
FKP density mesh.
"""
import logging

import numpy

from nbodykit_amd.source.catalog.species import MultipleSpeciesCatalog


def FKPWeightFromNbar(P0, nbar):
    """w_FKP = 1 / (1 + P0 nbar) (reference :7-27)."""
    if P0 != 0:
        return 1.0 / (1. + P0 * numpy.asarray(nbar))
    return 1.0


def get_data_bounds(pos, comm, selection=None):
    """min/max of the (selected) positions across ranks (the reference
    delegates to nbodykit/utils.py get_data_bounds)."""
    pos = numpy.asarray(pos)
    if selection is not None:
        pos = pos[numpy.asarray(selection, dtype=bool)]
    if len(pos):
        lo = pos.min(axis=0)
        hi = pos.max(axis=0)
    else:
        lo = numpy.full(3, numpy.inf)
        hi = numpy.full(3, -numpy.inf)
    lo = numpy.asarray(comm.allreduce(lo, op='min'))
    hi = numpy.asarray(comm.allreduce(hi, op='max'))
    return lo, hi


class FKPCatalog(MultipleSpeciesCatalog):
    logger = logging.getLogger('FKPCatalog')

    def __repr__(self):
        return "FKPCatalog(species=%s)" % str(self.attrs['species'])

    def __init__(self, data, randoms, BoxSize=None, BoxPad=0.02, P0=None,
                 nbar='NZ'):
        if randoms is None:
            randoms = data[:0]

        MultipleSpeciesCatalog.__init__(self, ['data', 'randoms'], data,
                                        randoms)

        for name in self.species:
            if nbar not in self[name]:
                raise ValueError("Column `%s` is not defined in `%s`"
                                 % (nbar, name))
        self.nbar = nbar

        for name in self.species:
            if P0 is not None:
                self[name]['FKPWeight'] = FKPWeightFromNbar(
                    P0, numpy.asarray(self[name][self.nbar]))
            elif 'FKPWeight' not in self[name]:
                self[name]['FKPWeight'] = 1.0

        if numpy.isscalar(BoxSize):
            BoxSize = numpy.ones(3) * BoxSize
        self.attrs['BoxSize'] = BoxSize
        if numpy.isscalar(BoxPad):
            BoxPad = numpy.ones(3) * BoxPad
        self.attrs['BoxPad'] = BoxPad

    def _define_bbox(self, position, selection, species):
        """BoxSize (padded, rounded up) and BoxCenter from the extent of
        ``species`` (reference :107-148)."""
        pos = numpy.asarray(self[species][position])
        sel = numpy.asarray(self[species][selection])
        pos_min, pos_max = get_data_bounds(pos, self.comm, selection=sel)

        if self.comm.rank == 0:
            self.logger.info("cartesian coordinate range: %s : %s"
                             % (str(pos_min), str(pos_max)))
        if numpy.isinf(pos_min).any() or numpy.isinf(pos_max).any():
            raise ValueError("Range of positions from `%s` is infinite; "
                             "try bbox_from_species='data'" % species)

        delta = abs(pos_max - pos_min)
        BoxCenter = 0.5 * (pos_min + pos_max)
        if self.attrs['BoxSize'] is None:
            delta = delta * (1.0 + self.attrs['BoxPad'])
            BoxSize = numpy.ceil(delta)
        else:
            BoxSize = self.attrs['BoxSize']
        return BoxSize, BoxCenter

    def to_mesh(self, Nmesh=None, BoxSize=None, BoxCenter=None,
                dtype='c16', interlaced=False, compensated=False,
                resampler='cic', fkp_weight='FKPWeight',
                comp_weight='Weight', selection='Selection',
                position='Position', bbox_from_species=None, window=None,
                nbar=None):
        from .catalogmesh import FKPCatalogMesh

        if window is not None:
            import warnings
            warnings.warn("the window argument is deprecated. Use "
                          "resampler= instead", DeprecationWarning)
            resampler = window

        for name in self.species:
            for col in [fkp_weight, comp_weight]:
                if col not in self[name]:
                    raise ValueError("the '%s' species is missing the "
                                     "'%s' column" % (name, col))

        if Nmesh is None:
            try:
                Nmesh = self.attrs['Nmesh']
            except KeyError:
                raise ValueError(
                    "cannot convert FKP source to a mesh; 'Nmesh' keyword "
                    "is not supplied and the FKP source does not define "
                    "one in 'attrs'.")

        if bbox_from_species is not None:
            BoxSize1, BoxCenter1 = self._define_bbox(position, selection,
                                                     bbox_from_species)
        else:
            if self['randoms'].csize > 0:
                BoxSize1, BoxCenter1 = self._define_bbox(position,
                                                         selection,
                                                         'randoms')
            else:
                BoxSize1, BoxCenter1 = self._define_bbox(position,
                                                         selection, 'data')

        if BoxSize is None:
            BoxSize = BoxSize1
        if BoxCenter is None:
            BoxCenter = BoxCenter1

        if self.comm.rank == 0:
            self.logger.info("BoxSize = %s" % str(BoxSize))
            self.logger.info("BoxCenter = %s" % str(BoxCenter))

        return FKPCatalogMesh(self, nbar=self.nbar,
                              comp_weight=comp_weight,
                              fkp_weight=fkp_weight, position=position,
                              value='Value', interlaced=interlaced,
                              compensated=compensated,
                              resampler=resampler, Nmesh=Nmesh,
                              BoxSize=BoxSize, BoxCenter=BoxCenter,
                              dtype=dtype, selection=selection)
