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


def _write_gadget1(fn, pos, vel, ids, masses, boxsize=100.,
                   pos_dtype='f4'):
    """Minimal Gadget-1 writer for the test: one ptype-1 block set with
    F77 markers, Massarr broadcast for the mass."""
    import struct
    n = len(pos)
    header = numpy.zeros(1, dtype=[
        ('Npart', ('u4', 6)), ('Massarr', ('f8', 6)), ('Time', 'f8'),
        ('Redshift', 'f8'), ('FlagSfr', 'i4'), ('FlagFeedback', 'i4'),
        ('Nall', ('u4', 6)), ('FlagCooling', 'i4'), ('NumFiles', 'i4'),
        ('BoxSize', 'f8'), ('Omega0', 'f8'), ('OmegaLambda', 'f8'),
        ('HubbleParam', 'f8'), ('FlagAge', 'i4'), ('FlagMetals', 'i4'),
        ('NallHW', ('u4', 6)), ('flag_entr_ics', 'i4')])[0]
    header['Npart'][1] = n
    header['Nall'][1] = n
    header['Massarr'][1] = masses
    header['BoxSize'] = boxsize
    header['Time'] = 1.0

    def block(ff, arr):
        raw = arr.tobytes()
        ff.write(struct.pack('i', len(raw)))
        ff.write(raw)
        ff.write(struct.pack('i', len(raw)))

    with open(fn, 'wb') as ff:
        raw = header.tobytes()
        raw += b'\0' * (256 - len(raw))
        ff.write(struct.pack('i', 256))
        ff.write(raw)
        ff.write(struct.pack('i', 256))
        block(ff, pos.astype(pos_dtype))
        block(ff, vel.astype('f4'))
        block(ff, ids.astype('i4'))


def test_gadget1_catalog(tmp_path):
    from nbodykit_amd.lab import Gadget1Catalog
    fn = str(tmp_path / 'snap')
    rng = numpy.random.RandomState(9)
    n = 64
    pos = rng.uniform(0, 100., size=(n, 3))
    vel = rng.normal(size=(n, 3))
    ids = numpy.arange(n)
    _write_gadget1(fn, pos, vel, ids, masses=0.125)
    cat = Gadget1Catalog(fn, ptype=1)
    assert cat.size == n
    nt.assert_allclose(numpy.asarray(cat['Position']),
                       pos.astype('f4'), rtol=1e-6)
    nt.assert_allclose(numpy.asarray(cat['GadgetVelocity']),
                       vel.astype('f4'), rtol=1e-6)
    nt.assert_array_equal(numpy.asarray(cat['ID']), ids)
    # Massarr broadcast
    nt.assert_allclose(numpy.asarray(cat['Mass']), 0.125)
    # header -> attrs
    assert cat.attrs['BoxSize'] == 100.
    assert cat.attrs['Time'] == 1.0


def test_gadget1_f8_positions(tmp_path):
    # float width inferred from the block size markers
    from nbodykit_amd.lab import Gadget1Catalog
    fn = str(tmp_path / 'snap8')
    rng = numpy.random.RandomState(10)
    n = 32
    pos = rng.uniform(0, 50., size=(n, 3))
    _write_gadget1(fn, pos, numpy.zeros((n, 3)), numpy.arange(n),
                   masses=1.0, pos_dtype='f8')
    cat = Gadget1Catalog(fn, ptype=1)
    assert numpy.asarray(cat['Position']).dtype == numpy.dtype('f8')
    nt.assert_allclose(numpy.asarray(cat['Position']), pos, rtol=1e-14)


def test_gadget1_bad_file(tmp_path):
    from nbodykit_amd.lab import Gadget1Catalog
    fn = str(tmp_path / 'junk')
    with open(fn, 'wb') as ff:
        ff.write(b'\x01\x02\x03\x04' * 100)
    with pytest.raises(IOError):
        Gadget1Catalog(fn)
