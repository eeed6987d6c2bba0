"""
CPU coverage of the product boundary: the C-ABI library loads and
exports every symbol include/nbk_hip.h declares; compute without a GPU
fails loudly (no silent fallback); the catalog column protocol, to_mesh
validation and generator reproducibility behave like the reference's.
"""
import os
import re

import numpy
import pytest
from numpy.testing import assert_allclose, assert_array_equal

import nbodykit_amd
from nbodykit_amd import hiplib, set_options, _global_options
from nbodykit_amd.lab import (UniformCatalog, LogNormalCatalog,
                              ArrayCatalog, FFTPower, LinearPower,
                              Planck15)

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


# ---- ABI ---------------------------------------------------------------

def test_library_loads_and_exports_header_symbols():
    lib = hiplib.load()
    header = open(os.path.join(HERE, 'include', 'nbk_hip.h')).read()
    declared = re.findall(r'\bint\s+(nbk_\w+)\s*\(', header)
    declared += re.findall(r'const char\*\s+(nbk_\w+)\s*\(', header)
    assert set(declared) == set(hiplib.EXPORTED_SYMBOLS)
    for sym in declared:
        assert hasattr(lib, sym), "missing export: %s" % sym
    assert b'gfx950' in lib.nbk_version()


def test_compute_fails_loudly_without_gpu():
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present")
    cat = UniformCatalog(nbar=1e-4, BoxSize=64., seed=42)
    mesh = cat.to_mesh(Nmesh=8)
    with pytest.raises(RuntimeError, match="no CPU fallback"):
        mesh.compute(mode='complex')
    with pytest.raises(RuntimeError, match="no CPU fallback"):
        FFTPower(cat, mode='1d', Nmesh=8)


# ---- catalog column protocol -------------------------------------------

def test_default_columns():
    cat = UniformCatalog(nbar=1e-4, BoxSize=64., seed=42)
    assert sorted(cat.columns) == ['Position', 'Selection', 'Value',
                                   'Velocity', 'Weight']
    assert_array_equal(numpy.asarray(cat['Selection']),
                       numpy.ones(cat.size, dtype=bool))
    assert_allclose(numpy.asarray(cat['Weight']), 1.0)
    assert_allclose(numpy.asarray(cat['Value']), 1.0)


def test_setitem_scalar_and_array():
    cat = UniformCatalog(nbar=1e-4, BoxSize=64., seed=42)
    cat['Weight'] = 2.0
    assert_allclose(numpy.asarray(cat['Weight']), 2.0)
    w = numpy.random.RandomState(1).uniform(size=cat.size)
    cat['Weight'] = w
    assert_array_equal(numpy.asarray(cat['Weight']), w)
    with pytest.raises(ValueError):
        cat['Weight'] = numpy.ones(cat.size + 1)
    with pytest.raises(KeyError):
        cat['Missing']


def test_slice_selection():
    cat = UniformCatalog(nbar=1e-4, BoxSize=64., seed=42)
    sub = cat[:0]
    assert sub.csize == 0
    mask = numpy.zeros(cat.size, dtype=bool)
    mask[:5] = True
    sub = cat[mask]
    assert sub.size == 5
    assert_array_equal(numpy.asarray(sub['Position']),
                       numpy.asarray(cat['Position'])[:5])


def test_to_mesh_validation():
    cat = UniformCatalog(nbar=1e-4, BoxSize=64., seed=42)
    with pytest.raises(ValueError, match="valid resampler"):
        cat.to_mesh(Nmesh=8, resampler='lanczos3')
    with pytest.raises(ValueError, match="Nmesh"):
        cat.to_mesh()
    mesh = cat.to_mesh(Nmesh=8)
    # reference defaults (base/catalog.py:787-790)
    assert mesh.dtype == 'f4'
    assert mesh.attrs['resampler'] == 'cic'
    assert mesh.attrs['compensated'] is False
    assert mesh.attrs['interlaced'] is False


def test_fftpower_validation():
    cat = UniformCatalog(nbar=1e-4, BoxSize=64., seed=42)
    with pytest.raises(ValueError, match="mode"):
        FFTPower(cat, mode='3d', Nmesh=8)
    with pytest.raises(ValueError, match="los"):
        FFTPower(cat, mode='1d', Nmesh=8, los=[0, 1])
    with pytest.raises(ValueError, match="los"):
        FFTPower(cat, mode='1d', Nmesh=8, los=[0, 0, 2])


def test_array_catalog():
    pos = numpy.random.RandomState(0).uniform(size=(10, 3)) * 10
    cat = ArrayCatalog({'Position': pos}, BoxSize=10.0)
    assert cat.size == 10
    assert_array_equal(numpy.asarray(cat['Position']), pos)
    with pytest.raises(ValueError):
        ArrayCatalog({'a': numpy.ones(3), 'b': numpy.ones(4)})


# ---- generator reproducibility -----------------------------------------

def test_uniform_catalog_reproducible():
    a = UniformCatalog(nbar=1e-4, BoxSize=64., seed=42)
    b = UniformCatalog(nbar=1e-4, BoxSize=64., seed=42)
    assert_array_equal(numpy.asarray(a['Position']),
                       numpy.asarray(b['Position']))
    c = UniformCatalog(nbar=1e-4, BoxSize=64., seed=43)
    assert not numpy.array_equal(numpy.asarray(a['Position']),
                                 numpy.asarray(c['Position']))


def test_uniform_matches_reference_recipe():
    """positions == serial-RandomState recipe of the reference
    (source/catalog/uniform.py:94-100) — bit-identical to upstream"""
    from tests.conftest import uniform_positions
    cat = UniformCatalog(nbar=3e-4, BoxSize=512., seed=42)
    want = uniform_positions(3e-4, 512., seed=42)
    assert_array_equal(numpy.asarray(cat['Position']), want)


def test_lognormal_reproducible_and_seeded():
    Plin = LinearPower(Planck15, redshift=0.55)
    kw = dict(Plin=Plin, nbar=1e-3, BoxSize=64., Nmesh=16, bias=2.0)
    a = LogNormalCatalog(seed=42, **kw)
    b = LogNormalCatalog(seed=42, **kw)
    assert_array_equal(numpy.asarray(a['Position']),
                       numpy.asarray(b['Position']))
    c = LogNormalCatalog(seed=43, **kw)
    assert a.csize != c.csize or not numpy.array_equal(
        numpy.asarray(a['Position']), numpy.asarray(c['Position']))
    # velocity columns exist and relate by the growth factor
    f = Planck15.scale_independent_growth_rate(0.55)
    voff = numpy.asarray(a['VelocityOffset'])
    assert voff.shape == (a.size, 3)


def test_set_options_paint_chunk_size():
    old = _global_options['paint_chunk_size']
    with set_options(paint_chunk_size=128):
        assert _global_options['paint_chunk_size'] == 128
    assert _global_options['paint_chunk_size'] == old
    with pytest.raises(KeyError):
        set_options(bogus=1)


def test_compensation_action_wiring():
    """compensated meshes must carry the compensation as their first
    action (source/mesh/catalog.py:405-451) — guards against the action
    pipeline silently dropping it (GPU parity then fails wholesale)."""
    from nbodykit_amd.source.mesh.catalog import (
        CompensateTSC, CompensateCICShotnoise, get_compensation,
        lookup_compensation)
    cat = UniformCatalog(nbar=1e-4, BoxSize=64., seed=42)
    mesh = cat.to_mesh(Nmesh=8, resampler='tsc', compensated=True,
                       interlaced=True)
    actions = mesh.actions
    assert len(actions) == 1
    assert actions[0] == ('complex', CompensateTSC, 'circular')

    mesh2 = cat.to_mesh(Nmesh=8, resampler='cic', compensated=True)
    assert mesh2.actions[0] == ('complex', CompensateCICShotnoise,
                                'circular')
    assert cat.to_mesh(Nmesh=8, compensated=False).actions == []

    # every builtin filter dispatches to the HIP kernel
    for interlaced in (False, True):
        for res in ('cic', 'tsc', 'pcs'):
            (mode, func, kind), = get_compensation(interlaced, res)
            assert mode == 'complex' and kind == 'circular'
            assert lookup_compensation(func) == (res, interlaced)


def test_catalog_copy():
    from nbodykit_amd.lab import UniformCatalog
    import numpy
    cat = UniformCatalog(nbar=1e-3, BoxSize=64., seed=1)
    cp = cat.copy()
    numpy.testing.assert_array_equal(numpy.asarray(cp['Position']),
                                     numpy.asarray(cat['Position']))
    assert cp.size == cat.size and cp.csize == cat.csize
    # attrs decoupled
    cp.attrs['extra'] = 1
    assert 'extra' not in cat.attrs
    # default columns still served
    assert bool(numpy.all(numpy.asarray(cp['Selection'])))


def test_catalog_gslice():
    from nbodykit_amd.lab import ArrayCatalog
    import numpy
    cat = ArrayCatalog({'Mass': numpy.arange(20.)})
    sl = cat.gslice(5, 15, 2)
    numpy.testing.assert_array_equal(numpy.asarray(sl['Mass']),
                                     numpy.arange(5., 15., 2.))
    assert sl.csize == 5


# ---- reference base/tests/test_catalog.py semantics ---------------------

def test_ref_slice_semantics():
    """reference test_slice (:230-250) + test_getitem KeyErrors."""
    import numpy
    from numpy.testing import assert_array_equal
    from nbodykit_amd.lab import UniformCatalog
    source = UniformCatalog(nbar=2e-4, BoxSize=512., seed=42)
    source['NZ'] = 1

    subset = source[:10]
    assert all(col in subset for col in source.columns)
    assert len(subset) == 10
    assert_array_equal(numpy.asarray(subset['Position']),
                       numpy.asarray(source['Position'])[:10])

    subset = source[[0, 1, 2]]
    assert_array_equal(numpy.asarray(subset['Position']),
                       numpy.asarray(source['Position'])[[0, 1, 2]])

    import pytest
    with pytest.raises(KeyError):
        source['BAD_COLUMN']


def test_ref_delitem_semantics():
    """reference test_delitem (:323-343)."""
    import numpy
    import pytest
    from nbodykit_amd.lab import UniformCatalog
    source = UniformCatalog(nbar=2e-4, BoxSize=512., seed=42)
    source['test'] = numpy.ones(source.size)
    with pytest.raises(ValueError):
        del source['Position']
    with pytest.raises(ValueError):
        del source['BAD_COLUMN']
    assert 'test' in source
    del source['test']
    assert 'test' not in source
    # a default column with no override is not deletable either
    with pytest.raises(ValueError):
        del source['Selection']
    # ... but becomes deletable once overridden
    source['Selection'] = numpy.ones(source.size, dtype=bool)
    del source['Selection']
    assert numpy.all(numpy.asarray(source['Selection']))


def test_ref_transform_arithmetic():
    """reference test_transform (:285-306): self-referential column
    reassignment resolves eagerly to the same values the reference's
    lazy graphs produce."""
    import numpy
    from numpy.testing import assert_allclose
    from nbodykit_amd.lab import ArrayCatalog
    data = numpy.ones(100, dtype=[('Position', ('f4', 3)),
                                  ('Velocity', ('f4', 3))])
    source = ArrayCatalog(data, BoxSize=100, Nmesh=32)
    source['Velocity'] = source['Position'] + source['Velocity']
    source['Position'] = source['Position'] + source['Velocity']
    assert_allclose(numpy.asarray(source['Position']), 3)
    mesh = source.to_mesh()
    numpy.testing.assert_array_equal(mesh.attrs['Nmesh'], 32)


def test_ref_column_masked_by_selection():
    """reference test_dask_slice (:255-268): slicing a column with a
    boolean column."""
    import numpy
    from numpy.testing import assert_array_equal
    from nbodykit_amd.lab import UniformCatalog
    source = UniformCatalog(nbar=2e-4, BoxSize=512., seed=42)
    index = numpy.random.RandomState(3).choice([True, False],
                                               size=len(source))
    source['Selection'] = index
    pos = numpy.asarray(source['Position'])
    pos2 = numpy.asarray(source['Position'])[
        numpy.asarray(source['Selection'], dtype=bool)]
    assert_array_equal(pos[index], pos2)


def test_lab_namespace_completeness():
    """The lab namespace carries every reference symbol this build
    supports (the drop-in checklist; reference nbodykit/lab.py)."""
    from nbodykit_amd import lab
    names = [
        # algorithms
        'FFTPower', 'ProjectedFFTPower', 'FFTCorr', 'FFTRecon',
        'ConvolvedFFTPower', 'RedshiftHistogram',
        # FKP
        'FKPCatalog', 'FKPWeightFromNbar',
        # catalogs
        'UniformCatalog', 'RandomCatalog', 'LogNormalCatalog',
        'ArrayCatalog', 'BigFileCatalog', 'CSVCatalog', 'BinaryCatalog',
        'Gadget1Catalog', 'MultipleSpeciesCatalog',
        # meshes
        'CatalogMesh', 'FieldMesh', 'ArrayMesh', 'LinearMesh',
        'BigFileMesh',
        # cosmology + helpers
        'LinearPower', 'Planck15', 'transform', 'filters',
        'BinnedStatistic', 'setup_logging', 'set_options',
        'CurrentMPIComm', 'cosmology', 'IO',
    ]
    missing = [n for n in names if not hasattr(lab, n)]
    assert not missing, "lab namespace missing: %s" % missing


def test_index_column():
    """reference test_index (:270-283): Index spans range(csize), i8,
    regenerated after gslice."""
    import numpy
    from nbodykit_amd.lab import UniformCatalog
    source = UniformCatalog(nbar=2e-4, BoxSize=512., seed=42)
    idx = source.Index
    numpy.testing.assert_array_equal(idx, numpy.arange(source.csize))
    assert idx.dtype == numpy.dtype('i8')
    sub = source.gslice(0, 100)
    numpy.testing.assert_array_equal(sub.Index, numpy.arange(sub.csize))


def test_mesh_setters_and_bad_window():
    """reference source/mesh/tests/test_catalogmesh.py:99-131."""
    import pytest
    from nbodykit_amd.lab import UniformCatalog
    source = UniformCatalog(nbar=3e-4, BoxSize=512., seed=42)
    mesh = source.to_mesh(resampler='cic', Nmesh=64, interlaced=True,
                          compensated=True)
    assert mesh.compensated is True
    mesh.compensated = False
    assert mesh.compensated is False
    assert mesh.interlaced is True
    mesh.interlaced = False
    assert mesh.interlaced is False
    assert mesh.window == 'cic'
    mesh.window = 'tsc'
    assert mesh.window == 'tsc'
    with pytest.raises(Exception):
        mesh.window = 'BAD'
    # unknown resampler at construction
    with pytest.raises(ValueError):
        source.to_mesh(resampler='db6', Nmesh=64)


def test_columnaccessor_semantics():
    """reference test_columnaccessor (:345-376): operations on a fetched
    column never write back to the catalog; explicit __setitem__ does."""
    import numpy
    from numpy.testing import assert_array_equal
    from nbodykit_amd.base.catalog import ColumnAccessor
    from nbodykit_amd.lab import UniformCatalog
    source = UniformCatalog(nbar=2e-4, BoxSize=512., seed=42)

    c = source['Position']
    truth = numpy.array(c[0])
    assert isinstance(c, ColumnAccessor)
    c *= 10.
    # c is no longer an accessor (it has transformed)...
    assert not isinstance(c, ColumnAccessor)
    # ...and the original is unaffected
    assert_array_equal(numpy.asarray(source['Position'])[0], truth)
    assert_array_equal(c[0], truth * 10.)

    # explicit in-place via __setitem__ works
    source['Position'] *= 10
    assert_array_equal(numpy.asarray(source['Position'])[0], truth * 10)

    # accessors carry their catalog
    new_col = source['Selection']
    assert isinstance(new_col, ColumnAccessor)
    source['Selection2'] = new_col
    assert source['Selection'].catalog is source


def test_catalog_view():
    """reference test_view (:416-440): shared columns + attrs,
    write-through."""
    from nbodykit_amd.lab import UniformCatalog
    source = UniformCatalog(nbar=2e-4, BoxSize=512., seed=42)
    source['TEST'] = 10.
    source.attrs['TEST'] = 10.0
    view = source.view()
    assert view.base is source
    assert isinstance(view, source.__class__)
    assert view.size == source.size and view.csize == source.csize
    for k in source.attrs:
        assert k in view.attrs
    view['TEST2'] = 5.0
    assert 'TEST2' in source
    source.attrs['foo'] = 123
    assert 'foo' in view.attrs
    import numpy
    numpy.testing.assert_array_equal(numpy.asarray(view['Position']),
                                     numpy.asarray(source['Position']))


def test_xbin_lds_gate_budget():
    """The Python LDS-budget gate for the deferred-x kernel must mirror
    the kernel's own check (a mismatch means a failed launch instead of
    a clean fallback)."""
    from nbodykit_amd.algorithms.fftpower import _xbin_lds_fits
    # C4 1d: 512 k-edges, Nmu=1 (2 mu edges) at n0=1024 fits
    assert _xbin_lds_fits(1024, 512, 2, 1)
    # 2d Nmu=5 at 1024^3 with ~512 edges does NOT (falls back)
    assert not _xbin_lds_fits(1024, 512, 6, 1)
    # interlaced pair needs two tiles + the phase table
    assert _xbin_lds_fits(512, 256, 2, 1, il=True)
    assert not _xbin_lds_fits(4096, 2048, 2, 1, il=True)


def test_pair_sort_gate_geometry():
    """Pair-bucket sort geometry picker: bucket count within the LDS
    ceiling, group == tile rows, graceful None for unsupported
    meshes."""
    from nbodykit_amd.source.mesh.catalog import _pair_gs

    assert _pair_gs([1024] * 3) == 4         # 16-row groups, 32768 buckets
    assert _pair_gs([512] * 3) == 5          # 32-row groups
    gs256 = _pair_gs([256] * 3)
    assert gs256 is not None and (256 >> 1) * (256 >> gs256) <= 40960
    assert _pair_gs([2048] * 3) is None      # beyond the LDS ceiling
    assert _pair_gs([27] * 3) is None        # odd n0 cannot pair


def test_mismatched_boxsize_raises():
    """Meshes with different BoxSize raise ValueError at FFTPower
    construction (reference test_fftpower.py:123-135) — checked before
    any compute, so it works without a GPU."""
    import pytest as _pytest
    from nbodykit_amd.lab import UniformCatalog, FFTPower
    m1 = UniformCatalog(nbar=1e-4, BoxSize=128., seed=1).to_mesh(Nmesh=16)
    m2 = UniformCatalog(nbar=1e-4, BoxSize=256., seed=1).to_mesh(Nmesh=16)
    with _pytest.raises(ValueError):
        FFTPower(m1, mode='1d', second=m2)


def test_redges_unique_matches_product():
    """Oracle's dr=0 unique-separation edges agree bit-for-bit with the
    product's host-side _find_unique_edges on the same coordinate grid
    (reference fftcorr.py:96-99 -> fftpower.py:732-769)."""
    import numpy
    from oracle.fftpower import redges_unique
    from oracle.mesh import MeshGeometry, real_coords
    from nbodykit_amd.algorithms.fftpower import _find_unique_edges

    class SerialComm:
        size = 1
        rank = 0

        def allgather(self, x):
            return [x]

        def allreduce(self, x, op=None):
            return x

    for nmesh, box in [(16, 100.), (12, 64.), (17, 80.)]:
        geom = MeshGeometry(nmesh, box, dtype='f8')
        rmax = 0.5 * box
        e1, c1 = redges_unique(geom, rmax)
        e2, c2 = _find_unique_edges(real_coords(geom),
                                    geom.BoxSize / geom.Nmesh,
                                    rmax, SerialComm())
        assert numpy.array_equal(e1, e2)
        assert numpy.array_equal(c1, c2)
