"""
Paint-driver restatement of ``CatalogMesh.to_real_field``
(nbodykit/source/mesh/catalog.py:155-403), single process:

chunked paint loop (chunk size = paint_chunk_size), optional interlacing
(two meshes half a cell apart combined in k-space at :340-354), particle
counters N/W/W2, shot noise V*W2/W^2 (:378) and the 1+delta normalization
(:394-398).
"""
import numpy

from .mesh import MeshGeometry, r2c, c2r, complex_coords
from .paint import paint


def to_real_field(position, geom, weight=None, value=None, resampler='cic',
                  interlaced=False, normalize=True,
                  paint_chunk_size=4 * 1024 * 1024):
    """
    Returns (mesh, attrs): mesh is the painted (optionally interlaced and
    1+delta-normalized) f8 array of shape geom.Nmesh; attrs carries
    shotnoise/N/W/W2/num_per_cell (reference :381-386).
    """
    position = numpy.asarray(position, dtype='f8')
    n = len(position)

    mesh = numpy.zeros(tuple(int(x) for x in geom.Nmesh), dtype='f8')
    if interlaced:
        mesh2 = numpy.zeros_like(mesh)

    if weight is None:
        weight = numpy.ones(n, dtype='f8')
    else:
        weight = numpy.asarray(weight, dtype='f8')
    if value is None:
        value = numpy.ones(n, dtype='f8')
    else:
        value = numpy.asarray(value, dtype='f8')

    N = n
    W = float(weight.sum())
    W2 = float((weight ** 2).sum())

    # chunked deposit, mirroring the driver loop at :303-332
    for start in range(0, max(n, 1), paint_chunk_size):
        sl = slice(start, start + paint_chunk_size)
        p = position[sl]
        m = weight[sl] * value[sl]
        paint(p, m, mesh, geom, resampler=resampler, shift=0.0)
        if interlaced:
            paint(p, m, mesh2, geom, resampler=resampler, shift=0.5)

    if interlaced:
        # combine: c = c1/2 + c2/2 * exp(0.5 i k.H) per mode (:341-347),
        # then back to configuration space (:351-354)
        c1 = r2c(mesh, geom)
        c2 = r2c(mesh2, geom)
        k = complex_coords(geom)
        H = geom.H
        kH = sum(k[i] * H[i] for i in range(3))
        c1 = c1 * 0.5 + c2 * 0.5 * numpy.exp(0.5 * 1j * kH)
        mesh = c2r(c1, geom)

    nbar = 1.0 * W / float(numpy.prod(geom.Nmesh))

    with numpy.errstate(divide='ignore', invalid='ignore'):
        shotnoise = float(numpy.prod(geom.BoxSize)) * W2 / W ** 2 \
            if W != 0 else numpy.nan

    attrs = {
        'shotnoise': shotnoise,
        'N': N,
        'W': W,
        'W2': W2,
        'num_per_cell': nbar,
    }

    if normalize:
        if nbar > 0:
            mesh /= nbar
        else:
            mesh[...] = 1.0

    return mesh, attrs
