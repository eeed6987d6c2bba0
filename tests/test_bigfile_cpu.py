"""
CPU tests for the bigfile on-disk format restatement
(nbodykit_amd/io/bigfile_format.py) and the catalog save/load path —
behavioral parity with the reference's io/tests/test_bigfile.py (which
generates files at run time with the external ``bigfile`` package; no
committed fixtures exist, so byte-level parity vs that library is
unpinned and documented as such in the module docstring).
"""
import os

import numpy
import numpy.testing as nt
import pytest

from nbodykit_amd.io.bigfile_format import BigFile, Block, Dataset
from nbodykit_amd.lab import ArrayCatalog, BigFileCatalog


# --------------------------------------------------------------- format
def test_block_roundtrip(tmp_path):
    path = str(tmp_path / 'bf')
    data = numpy.random.RandomState(0).random_sample((1024, 3))
    with BigFile(path, create=True) as ff:
        with ff.create('Position', dtype=('f4', 3), size=1024) as bb:
            bb.write(0, data)   # f8 cast to f4 on write (reference
                                # io/tests/test_bigfile.py:21-23)
    ff = BigFile(path)
    bb = ff['Position']
    assert bb.size == 1024 and bb.nmemb == 3
    assert bb.dtype == numpy.dtype('<f4')
    nt.assert_array_equal(bb[:], data.astype('f4'))
    nt.assert_array_equal(bb[10:20], data[10:20].astype('f4'))
    nt.assert_array_equal(bb[::2], data[::2].astype('f4'))


def test_block_multi_file(tmp_path):
    # rows split across NFILE physical files; reads cross boundaries
    path = str(tmp_path / 'bf')
    data = numpy.arange(100, dtype='i8')
    with BigFile(path, create=True) as ff:
        with ff.create('x', dtype='i8', size=100, Nfile=3) as bb:
            bb.write(0, data)
    files = sorted(os.listdir(os.path.join(path, 'x')))
    assert files == ['000000', '000001', '000002', 'header']
    bb = BigFile(path)['x']
    nt.assert_array_equal(bb[:], data)
    nt.assert_array_equal(bb[30:40], data[30:40])
    # header is the documented text layout
    head = open(os.path.join(path, 'x', 'header')).read().splitlines()
    assert head[0] == 'DTYPE: <i8'
    assert head[1] == 'NMEMB: 1'
    assert head[2] == 'NFILE: 3'
    assert head[3].startswith('000000: 34 :')


def test_block_offset_writes(tmp_path):
    path = str(tmp_path / 'bf')
    data = numpy.arange(50, dtype='f8')
    with BigFile(path, create=True) as ff:
        with ff.create('x', dtype='f8', size=50, Nfile=2) as bb:
            bb.write(30, data[30:])
            bb.write(0, data[:30])
    nt.assert_array_equal(BigFile(path)['x'][:], data)


def test_attrs_roundtrip(tmp_path):
    path = str(tmp_path / 'bf')
    with BigFile(path, create=True) as ff:
        with ff.create('Header') as bb:       # attrs-only block
            bb.attrs['Size'] = 1024
            bb.attrs['BoxSize'] = numpy.array([1.0, 2.0, 3.0])
            bb.attrs['name'] = 'json://["a", 1]'
            bb.attrs['flag'] = True
    bb = BigFile(path)['Header']
    assert bb.attrs['Size'] == 1024
    nt.assert_array_equal(bb.attrs['BoxSize'], [1.0, 2.0, 3.0])
    assert bb.attrs['name'] == 'json://["a", 1]'
    assert bb.attrs['flag'] == 1
    assert 'Header' in BigFile(path).blocks


def test_nested_dataset(tmp_path):
    # reference io/tests/test_bigfile.py:30-60: dataset '1' with its
    # own '.' attrs block
    path = str(tmp_path / 'bf')
    data = numpy.random.RandomState(1).random_sample((64, 3))
    with BigFile(path, create=True) as ff:
        with ff.create('1/.') as bb:
            bb.attrs['Over'] = 64
        with ff.create('1/Position', dtype=('f4', 3), size=64) as bb:
            bb.write(0, data)
    ff = BigFile(path)
    assert '1/Position' in ff.blocks and '1' in ff.blocks
    sub = ff['1/']
    assert 'Position' in sub.blocks
    ds = Dataset(sub, ['Position'])
    assert ds.size == 64
    assert ds.dtype == numpy.dtype([('Position', ('<f4', (3,)))])
    nt.assert_array_equal(ds[0:64]['Position'], data.astype('f4'))
    assert ff['1/.'].attrs['Over'] == 64


def test_dataset_unequal_sizes(tmp_path):
    path = str(tmp_path / 'bf')
    with BigFile(path, create=True) as ff:
        with ff.create('a', dtype='f8', size=10) as bb:
            bb.write(0, numpy.zeros(10))
        with ff.create('b', dtype='f8', size=11) as bb:
            bb.write(0, numpy.zeros(11))
    with pytest.raises(ValueError):
        Dataset(BigFile(path)['./'], ['a', 'b'])


# --------------------------------------------------------------- catalog
def _catalog(n=333, seed=3):
    rng = numpy.random.RandomState(seed)
    cat = ArrayCatalog({'Position': rng.uniform(0, 100., size=(n, 3)),
                        'Mass': rng.exponential(size=n)})
    cat.attrs['BoxSize'] = numpy.array([100., 100., 100.])
    cat.attrs['note'] = {'kind': 'test'}     # forces json:// encoding
    return cat


def test_catalog_save_load(tmp_path):
    path = str(tmp_path / 'cat')
    cat = _catalog()
    cat.save(path)
    loaded = BigFileCatalog(path)
    assert loaded.size == cat.size and loaded.csize == cat.csize
    nt.assert_array_equal(loaded['Position'], cat['Position'])
    nt.assert_array_equal(loaded['Mass'], cat['Mass'])
    nt.assert_array_equal(loaded.attrs['BoxSize'], [100.] * 3)
    assert loaded.attrs['note'] == {'kind': 'test'}
    # default columns not written, but still available
    assert not os.path.exists(os.path.join(path, 'Selection'))
    assert bool(numpy.all(loaded['Selection']))
    # and the loaded catalog converts to a mesh view
    mesh = loaded.to_mesh(Nmesh=8)
    nt.assert_array_equal(mesh.attrs['Nmesh'], 8)


def test_catalog_save_dataset_prefix(tmp_path):
    path = str(tmp_path / 'cat')
    cat = _catalog(64)
    cat.save(path, columns=['Position'], dataset='1')
    loaded = BigFileCatalog(path, dataset='1')
    nt.assert_array_equal(loaded['Position'], cat['Position'])
    # header at the root still read
    nt.assert_array_equal(loaded.attrs['BoxSize'], [100.] * 3)


def test_catalog_save_big_columns_split(tmp_path):
    # >32Mi rows per file forces NFILE>1; use a tiny override through
    # the format API instead of 32M rows
    path = str(tmp_path / 'bf')
    rows = numpy.arange(70_000_000, dtype='i1')  # 70M i1 rows, 3 files
    with BigFile(path, create=True) as ff:
        bb = ff.create_from_array('x', rows)
    assert len(bb._file_sizes) == 3
    got = BigFile(path)['x']
    assert got.size == len(rows)
    nt.assert_array_equal(got[:100], rows[:100])
    nt.assert_array_equal(got[40_000_000:40_000_100],
                          rows[40_000_000:40_000_100])


def test_bigfilemesh_attrs_cpu(tmp_path):
    # constructing BigFileMesh needs no GPU: attrs + pm metadata only
    from nbodykit_amd.lab import BigFileMesh
    path = str(tmp_path / 'mesh')
    with BigFile(path, create=True) as ff:
        with ff.create('Field', dtype='f8', size=8 ** 3) as bb:
            bb.write(0, numpy.zeros(8 ** 3))
            bb.attrs['ndarray.shape'] = numpy.array([8, 8, 8])
            bb.attrs['Nmesh'] = numpy.array([8, 8, 8])
            bb.attrs['BoxSize'] = numpy.array([1., 1., 1.])
            bb.attrs['extra'] = 'json://{"a": 1}'
    mesh = BigFileMesh(path, 'Field')
    assert not mesh.isfourier
    nt.assert_array_equal(mesh.attrs['Nmesh'], 8)
    assert mesh.attrs['extra'] == {'a': 1}
    with BigFile(path, create=True) as ff:
        with ff.create('FieldC', dtype='c16', size=8 * 8 * 5) as bb:
            bb.write(0, numpy.zeros(8 * 8 * 5, dtype='c16'))
            bb.attrs['ndarray.shape'] = numpy.array([8, 8, 5])
            bb.attrs['Nmesh'] = numpy.array([8, 8, 8])
            bb.attrs['BoxSize'] = numpy.array([1., 1., 1.])
    meshc = BigFileMesh(path, 'FieldC')
    assert meshc.isfourier and meshc.dtype == 'f8'
