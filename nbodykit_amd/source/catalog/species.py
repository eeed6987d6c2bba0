"""
MultipleSpeciesCatalog — a named collection of CatalogSources sharing one
communicator (reference nbodykit/source/catalog/species.py, the subset
the FKP pipeline uses: the ``species`` list, ``self[name]`` access to
the underlying catalogs, and merged attrs with ``name.`` prefixes).
"""
import numpy

from nbodykit_amd.base.catalog import CatalogSourceBase


class MultipleSpeciesCatalog(CatalogSourceBase):

    def __repr__(self):
        return "MultipleSpeciesCatalog(species=%s)" \
            % str(self.attrs['species'])

    def __init__(self, names, *species, **kwargs):
        if len(set(names)) != len(names):
            raise ValueError("species names must be unique")
        if len(names) != len(species):
            raise ValueError("a name is required for each species catalog "
                             "provided")

        self.comm = species[0].comm
        for cat in species:
            if cat.comm is not self.comm:
                raise ValueError("communicator mismatch between species")

        self.attrs['species'] = list(names)
        self._species = dict(zip(names, species))

        # store the species attrs with prefixed keys (reference :88-92)
        for name, cat in zip(names, species):
            for key, value in cat.attrs.items():
                self.attrs['%s.%s' % (name, key)] = value

    @property
    def species(self):
        return self.attrs['species']

    def __getitem__(self, key):
        if isinstance(key, str) and key in self._species:
            return self._species[key]
        if isinstance(key, str) and '/' in key:
            name, col = key.split('/', 1)
            return self._species[name][col]
        raise KeyError("column access must be 'species' or "
                       "'species/column'; got %r" % (key,))

    def __setitem__(self, key, value):
        if isinstance(key, str) and '/' in key:
            name, col = key.split('/', 1)
            self._species[name][col] = value
            return
        raise KeyError("set columns as 'species/column'")

    def to_mesh(self, Nmesh=None, BoxSize=None, dtype='f4',
                interlaced=False, compensated=False, resampler='cic',
                weight='Weight', value='Value', selection='Selection',
                position='Position', window=None):
        """Mesh painting the summed density of all species (reference
        :157-224)."""
        from nbodykit_amd.source.mesh.species import \
            MultipleSpeciesCatalogMesh
        if window is not None:
            raise RuntimeError("use resampler instead")
        for name in self.species:
            for col in [position, selection, weight, value]:
                if col not in self[name]:
                    raise ValueError("the '%s' species is missing the "
                                     "'%s' column" % (name, col))
        if BoxSize is None:
            BoxSize = _species_metadata('BoxSize', self.attrs,
                                        self.species)
        if Nmesh is None:
            Nmesh = _species_metadata('Nmesh', self.attrs, self.species)
        return MultipleSpeciesCatalogMesh(
            self, Nmesh=Nmesh, BoxSize=BoxSize, dtype=dtype,
            selection=selection, position=position, weight=weight,
            value=value, interlaced=interlaced, compensated=compensated,
            resampler=resampler)

    def __contains__(self, key):
        if key in self._species:
            return True
        if isinstance(key, str) and '/' in key:
            name, col = key.split('/', 1)
            return name in self._species and col in self._species[name]
        return False


def _species_metadata(name, attrs, species):
    """The single value of ``name`` across all species' attrs
    (reference check_species_metadata, :227-252)."""
    vals = []
    if name in attrs:
        vals.append(attrs[name])
    for s in species:
        key = '%s.%s' % (s, name)
        if key in attrs:
            vals.append(attrs[key])
    if not vals:
        raise ValueError("please specify %s — it is not defined in the "
                         "species metadata" % name)
    first = numpy.asarray(vals[0])
    for v in vals[1:]:
        if not numpy.array_equal(first, numpy.asarray(v)):
            raise ValueError("%s is inconsistent between species" % name)
    return vals[0]
