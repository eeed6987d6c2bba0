"""
Column transformations (reference nbodykit/transform.py — the numpy
restatement: the reference operates on dask arrays, which are an
implementation detail here; columns are numpy-backed either way).

``frame='galactic'`` variants need astropy (absent in this container)
and raise NotImplementedError.
"""
import numpy

from nbodykit_amd.base.catalog import ConstantArray   # noqa: F401

C_KMS = 299792.458      # speed of light in km/s (astropy c.to('km/s'))


def StackColumns(*cols):
    """Stack 1d columns as a (N, ncol) array (reference :5-27)."""
    cols = [numpy.asarray(c) for c in cols]
    for c in cols:
        if c.ndim != 1:
            raise TypeError("all input columns must be 1d arrays")
    return numpy.vstack(cols).T


def ConcatenateSources(*sources, **kwargs):
    """Concatenate CatalogSource objects together row-wise (reference
    :29-87): every hard/overridden column present in all sources is
    stacked; attrs merge with later sources winning."""
    from nbodykit_amd.source.catalog.array import ArrayCatalog

    columns = kwargs.get('columns', None)
    if columns is None:
        columns = set.intersection(*[
            set(c for c in src.columns if not src[c].is_default)
            for src in sources])
        columns = sorted(columns)
    elif isinstance(columns, str):
        columns = [columns]

    data = {col: numpy.concatenate([numpy.asarray(src[col])
                                    for src in sources], axis=0)
            for col in columns}
    toret = ArrayCatalog(data, comm=sources[0].comm)
    for src in sources:
        toret.attrs.update(src.attrs)
    return toret


def CartesianToEquatorial(pos, observer=[0, 0, 0], frame='icrs'):
    """(ra, dec) in degrees from Cartesian positions (reference
    :110-177); ra in [0, 360), dec in [-90, 90]."""
    if frame != 'icrs':
        raise NotImplementedError(
            "frame=%r needs astropy, which is not available" % frame)
    pos = numpy.asarray(pos)
    x, y, z = (pos[..., i] - observer[i] for i in range(3))
    s = numpy.hypot(x, y)
    ra = numpy.mod(numpy.rad2deg(numpy.arctan2(y, x)) - 360., 360.)
    dec = numpy.rad2deg(numpy.arctan2(z, s))
    return numpy.stack((ra, dec), axis=0)


def CartesianToSky(pos, cosmo, velocity=None, observer=[0, 0, 0],
                   zmax=100., frame='icrs'):
    """(ra, dec, z) from Cartesian position (+ optional peculiar
    velocity redshift offset) (reference :179-265)."""
    from scipy.interpolate import interp1d
    pos = numpy.asarray(pos) - numpy.asarray(observer)
    ra, dec = CartesianToEquatorial(pos, frame=frame)

    r = numpy.linalg.norm(pos, axis=-1)

    zgrid = numpy.concatenate([[0.],
                               numpy.logspace(-8, numpy.log10(zmax), 1024)])
    rgrid = cosmo.comoving_distance(zgrid)
    z = interp1d(rgrid, zgrid)(r)

    if velocity is not None:
        velocity = numpy.asarray(velocity)
        vpec = (pos * velocity).sum(axis=-1) / r
        z = z + vpec / C_KMS * (1 + z)

    return numpy.stack((ra, dec, z), axis=0)


def SkyToUnitSphere(ra, dec, degrees=True, frame='icrs'):
    """Unit-sphere Cartesian coordinates from (ra, dec) (reference
    :266-330)."""
    if frame != 'icrs':
        raise NotImplementedError(
            "frame=%r needs astropy, which is not available" % frame)
    ra, dec = numpy.broadcast_arrays(numpy.asarray(ra, dtype='f8'),
                                     numpy.asarray(dec, dtype='f8'))
    if degrees:
        ra = numpy.deg2rad(ra)
        dec = numpy.deg2rad(dec)
    x = numpy.cos(dec) * numpy.cos(ra)
    y = numpy.cos(dec) * numpy.sin(ra)
    z = numpy.sin(dec)
    return numpy.vstack([x, y, z]).T


def SkyToCartesian(ra, dec, redshift, cosmo, observer=[0, 0, 0],
                   degrees=True, frame='icrs'):
    """Cartesian Position column in Mpc/h from (ra, dec, z) (reference
    :331-374)."""
    ra, dec, redshift = numpy.broadcast_arrays(
        numpy.asarray(ra, dtype='f8'), numpy.asarray(dec, dtype='f8'),
        numpy.asarray(redshift, dtype='f8'))
    pos = SkyToUnitSphere(ra, dec, degrees=degrees, frame=frame)
    r = cosmo.comoving_distance(redshift)
    return r[:, None] * pos + numpy.asarray(observer)


def VectorProjection(vector, direction):
    """(v . dhat) dhat (reference :489-530)."""
    vector = numpy.asarray(vector)
    direction = numpy.asarray(direction, dtype='f8')
    direction = direction / (direction ** 2).sum(axis=-1,
                                                 keepdims=True) ** 0.5
    return (vector * direction).sum(axis=-1)[:, None] * direction
