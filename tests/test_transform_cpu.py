"""CPU tests for the transform module (reference nbodykit/transform.py)
and Cosmology.comoving_distance."""
import numpy
import numpy.testing as nt
import pytest

from nbodykit_amd.lab import transform, ArrayCatalog
from nbodykit_amd.cosmology import Planck15


def test_comoving_distance():
    # Planck15 flat LCDM: known values (Mpc/h)
    nt.assert_allclose(Planck15.comoving_distance(0.0), 0.0, atol=1e-10)
    nt.assert_allclose(Planck15.comoving_distance(1.0), 2300.6, rtol=2e-3)
    nt.assert_allclose(Planck15.comoving_distance(0.5), 1322.0, rtol=5e-3)
    # monotone and vectorized
    z = numpy.linspace(0, 3, 50)
    d = Planck15.comoving_distance(z)
    assert (numpy.diff(d) > 0).all()


def test_sky_to_unit_sphere():
    ra = numpy.array([0., 90., 180.])
    dec = numpy.array([0., 0., 90.])
    pos = transform.SkyToUnitSphere(ra, dec)
    nt.assert_allclose(pos, [[1, 0, 0], [0, 1, 0], [0, 0, 1]], atol=1e-12)
    nt.assert_allclose(numpy.linalg.norm(pos, axis=-1), 1.0)
    with pytest.raises(NotImplementedError):
        transform.SkyToUnitSphere(ra, dec, frame='galactic')


def test_sky_cartesian_roundtrip():
    rng = numpy.random.RandomState(3)
    ra = rng.uniform(0, 360., 200)
    dec = rng.uniform(-89., 89., 200)
    z = rng.uniform(0.1, 1.5, 200)
    pos = transform.SkyToCartesian(ra, dec, z, Planck15)
    ra2, dec2, z2 = transform.CartesianToSky(pos, Planck15)
    nt.assert_allclose(ra2, ra, rtol=1e-9, atol=1e-9)
    nt.assert_allclose(dec2, dec, rtol=1e-9, atol=1e-9)
    # the z(r) inversion uses the reference's 1024-point log grid to
    # zmax (transform.py:229-233): ~3e-4 interpolation accuracy
    nt.assert_allclose(z2, z, rtol=5e-4)


def test_cartesian_to_sky_with_velocity():
    pos = numpy.array([[100., 0., 0.]])
    vel = numpy.array([[300., 0., 0.]])        # radial, km/s
    _, _, z0 = transform.CartesianToSky(pos, Planck15)
    _, _, z1 = transform.CartesianToSky(pos, Planck15, velocity=vel)
    expect = z0 + 300. / transform.C_KMS * (1 + z0)
    nt.assert_allclose(z1, expect, rtol=1e-12)


def test_stack_and_projection():
    a = numpy.arange(4.)
    b = numpy.arange(4.) + 10
    st = transform.StackColumns(a, b)
    assert st.shape == (4, 2)
    nt.assert_array_equal(st[:, 1], b)
    with pytest.raises(TypeError):
        transform.StackColumns(st, a)

    v = numpy.array([[1., 1., 0.], [0., 2., 0.]])
    d = numpy.array([[1., 0., 0.], [0., 1., 0.]])
    proj = transform.VectorProjection(v, d)
    nt.assert_allclose(proj, [[1, 0, 0], [0, 2, 0]], atol=1e-13)


def test_concatenate_sources():
    c1 = ArrayCatalog({'Position': numpy.ones((5, 3)),
                       'Mass': numpy.arange(5.)})
    c2 = ArrayCatalog({'Position': numpy.zeros((3, 3)),
                       'Mass': numpy.arange(3.) + 10})
    cat = transform.ConcatenateSources(c1, c2)
    assert cat.size == 8
    nt.assert_array_equal(numpy.asarray(cat['Mass']),
                          numpy.concatenate([numpy.arange(5.),
                                             numpy.arange(3.) + 10]))
    nt.assert_array_equal(numpy.asarray(cat['Position'])[:5], 1.0)
