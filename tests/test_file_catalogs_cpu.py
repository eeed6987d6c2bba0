"""CPU tests for BinaryCatalog / CSVCatalog (reference io/binary.py,
io/csv.py behavior: column-major binary layout, headerless CSV with
names, dtype casting, usecols)."""
import numpy
import numpy.testing as nt
import pytest

from nbodykit_amd.lab import BinaryCatalog, CSVCatalog


def test_binary_catalog(tmp_path):
    fn = str(tmp_path / 'cat.bin')
    pos = numpy.random.RandomState(0).random_sample((128, 3))
    vel = numpy.random.RandomState(1).random_sample((128, 3))
    # column-major: all of Position then all of Velocity
    with open(fn, 'wb') as ff:
        pos.tofile(ff)
        vel.tofile(ff)
    cat = BinaryCatalog(fn, [('Position', ('f8', 3)),
                             ('Velocity', ('f8', 3))])
    assert cat.size == 128 and cat.csize == 128
    nt.assert_array_equal(numpy.asarray(cat['Position']), pos)
    nt.assert_array_equal(numpy.asarray(cat['Velocity']), vel)
    # size mismatch detection
    with open(fn, 'ab') as ff:
        ff.write(b'x')
    with pytest.raises(ValueError):
        BinaryCatalog(fn, [('Position', ('f8', 3)),
                           ('Velocity', ('f8', 3))])


def test_binary_catalog_header_and_scalar(tmp_path):
    fn = str(tmp_path / 'cat.bin')
    mass = numpy.arange(50, dtype='f4')
    with open(fn, 'wb') as ff:
        ff.write(b'\0' * 16)            # header
        mass.tofile(ff)
    cat = BinaryCatalog(fn, [('Mass', 'f4')], header_size=16)
    assert cat.size == 50
    nt.assert_array_equal(numpy.asarray(cat['Mass']), mass)


def test_csv_catalog(tmp_path):
    fn = str(tmp_path / 'cat.txt')
    rng = numpy.random.RandomState(2)
    data = rng.random_sample((40, 5))
    numpy.savetxt(fn, data)
    names = ['a', 'b', 'c', 'd', 'e']
    cat = CSVCatalog(fn, names)
    assert cat.size == 40
    for i, n in enumerate(names):
        nt.assert_allclose(numpy.asarray(cat[n]), data[:, i])
    # usecols + dtype cast
    cat2 = CSVCatalog(fn, names, usecols=['b', 'd'], dtype='f4')
    assert sorted(c for c in cat2.columns
                  if c not in ('Selection', 'Weight', 'Value')) \
        == ['b', 'd']
    assert numpy.asarray(cat2['b']).dtype == numpy.dtype('f4')


def test_csv_catalog_to_mesh(tmp_path):
    # stacked columns feed the mesh path
    from nbodykit_amd.lab import transform
    fn = str(tmp_path / 'cat.txt')
    rng = numpy.random.RandomState(3)
    numpy.savetxt(fn, rng.uniform(0, 32., size=(100, 3)))
    cat = CSVCatalog(fn, ['x', 'y', 'z'])
    cat['Position'] = transform.StackColumns(cat['x'], cat['y'],
                                             cat['z'])
    mesh = cat.to_mesh(Nmesh=16, BoxSize=32.)
    numpy.testing.assert_array_equal(mesh.attrs['Nmesh'], 16)


def test_csv_catalog_glob(tmp_path):
    rng = numpy.random.RandomState(5)
    chunks = []
    for i in range(3):
        d = rng.random_sample((10, 2))
        numpy.savetxt(str(tmp_path / ('part%d.txt' % i)), d)
        chunks.append(d)
    cat = CSVCatalog(str(tmp_path / 'part*.txt'), ['x', 'y'])
    assert cat.size == 30
    nt.assert_allclose(numpy.asarray(cat['x']),
                       numpy.concatenate(chunks)[:, 0])


def test_binary_catalog_multifile(tmp_path):
    fns = []
    parts = []
    for i in range(2):
        fn = str(tmp_path / ('b%d.bin' % i))
        d = numpy.arange(8, dtype='f8') + 100 * i
        d.tofile(fn)
        fns.append(fn)
        parts.append(d)
    cat = BinaryCatalog(fns, [('Mass', 'f8')])
    assert cat.size == 16
    nt.assert_array_equal(numpy.asarray(cat['Mass']),
                          numpy.concatenate(parts))
