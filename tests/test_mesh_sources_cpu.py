"""CPU-side tests for ArrayMesh / LinearMesh (reference
source/mesh/array.py, linear.py): construction, attrs, host-side
complex conversion.  Compute paths are GPU tests."""
import numpy
import numpy.testing as nt
import pytest

from nbodykit_amd.lab import ArrayMesh, LinearMesh


def test_arraymesh_attrs():
    arr = numpy.arange(4 ** 3, dtype='f8').reshape(4, 4, 4)
    mesh = ArrayMesh(arr, BoxSize=10., note='x')
    nt.assert_array_equal(mesh.attrs['Nmesh'], 4)
    nt.assert_array_equal(mesh.attrs['BoxSize'], 10.)
    assert mesh.attrs['note'] == 'x'
    nt.assert_array_equal(mesh._array, arr)


def test_arraymesh_complex_input():
    # a complex array is taken to configuration space on the host with
    # the reference's normalization (irfftn * Ntot inverts r2c/N^3)
    rng = numpy.random.RandomState(0)
    arr = rng.random_sample((8, 8, 8))
    cplx = numpy.fft.rfftn(arr) / arr.size
    mesh = ArrayMesh(cplx, BoxSize=1.)
    nt.assert_allclose(mesh._array, arr, rtol=1e-12, atol=1e-13)


def test_arraymesh_rejects_2d():
    with pytest.raises(ValueError):
        ArrayMesh(numpy.zeros((4, 4)), BoxSize=1.)


def test_linearmesh_attrs_and_seed():
    P = lambda k: numpy.ones_like(k)
    mesh = LinearMesh(P, BoxSize=100., Nmesh=16, seed=7)
    assert mesh.attrs['seed'] == 7
    assert mesh.attrs['unitary_amplitude'] is False
    # deprecated remove_variance maps onto unitary_amplitude
    mesh2 = LinearMesh(P, BoxSize=100., Nmesh=16, seed=7,
                       remove_variance=True)
    assert mesh2.attrs['unitary_amplitude'] is True
    # auto seed draws and broadcasts
    mesh3 = LinearMesh(P, BoxSize=100., Nmesh=16)
    assert 0 <= mesh3.attrs['seed'] < 4294967295
