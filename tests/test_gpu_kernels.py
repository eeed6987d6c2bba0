"""
Kernel-level GPU parity: each HIP kernel against its oracle restatement
on the same seeded inputs.  f64 bars: FFT/elementwise ~1e-12 relative
(pure fp ordering differences), paint 1e-12 (atomic order only).
"""
import numpy
import pytest
from numpy.testing import assert_allclose

pytestmark = pytest.mark.gpu

torch = pytest.importorskip('torch')
if not torch.cuda.is_available():
    pytest.skip('no GPU', allow_module_level=True)

from nbodykit_amd import hiplib                                 # noqa: E402
from oracle import (MeshGeometry, r2c, c2r, paint,              # noqa: E402
                    apply_compensation, compute_3d_power)
from oracle.mesh import complex_circular_coords                 # noqa: E402


def dev(arr):
    return torch.as_tensor(numpy.ascontiguousarray(arr)).to('cuda')


def host(t):
    torch.cuda.synchronize()
    return t.cpu().numpy()


@pytest.fixture(scope='module')
def lib():
    return hiplib.require()


def rand_positions(n, box, seed=1):
    rng = numpy.random.RandomState(seed)
    return rng.uniform(0, box, size=(n, 3))


# ---- paint --------------------------------------------------------------

@pytest.mark.parametrize('window', ['cic', 'tsc', 'pcs'])
@pytest.mark.parametrize('shift', [0.0, 0.5])
def test_paint_matches_oracle(lib, window, shift):
    geom = MeshGeometry(16, 32.)
    pos = rand_positions(5000, 32., seed=3)
    mass = numpy.random.RandomState(4).uniform(0.5, 1.5, size=len(pos))

    want = numpy.zeros((16, 16, 16))
    paint(pos, mass, want, geom, resampler=window, shift=shift)

    mesh_t = torch.zeros((16, 16, 16), dtype=torch.float64, device='cuda')
    pos_soa = dev(pos).t().contiguous()
    mass_t = dev(mass)     # keep alive: the kernel holds only the pointer
    hiplib.check(lib.nbk_paint_f64(
        hiplib.dptr(pos_soa), hiplib.dptr(mass_t), len(pos),
        hiplib.i64_arr(geom.Nmesh), hiplib.f64_arr(geom.BoxSize),
        hiplib.WINDOW_IDS[window], shift,
        hiplib.dptr(mesh_t), 0, 16, None), 'paint')
    got = host(mesh_t)
    assert_allclose(got, want, rtol=1e-12, atol=1e-12)
    # total mass deposited == sum of masses
    assert_allclose(got.sum(), mass.sum(), rtol=1e-12)


def test_paint_unit_mass_null_pointer(lib):
    geom = MeshGeometry(8, 8.)
    pos = rand_positions(100, 8., seed=5)
    want = numpy.zeros((8, 8, 8))
    paint(pos, 1.0, want, geom, resampler='cic')
    mesh_t = torch.zeros((8, 8, 8), dtype=torch.float64, device='cuda')
    pos_soa = dev(pos).t().contiguous()
    hiplib.check(lib.nbk_paint_f64(
        hiplib.dptr(pos_soa), None, len(pos),
        hiplib.i64_arr(geom.Nmesh), hiplib.f64_arr(geom.BoxSize),
        0, 0.0, hiplib.dptr(mesh_t), 0, 8, None), 'paint')
    assert_allclose(host(mesh_t), want, rtol=1e-12, atol=1e-12)


def test_paint_slab_bounds(lib):
    """painting into two half-slabs == painting the full mesh"""
    geom = MeshGeometry(16, 32.)
    pos = rand_positions(2000, 32., seed=6)
    full = numpy.zeros((16, 16, 16))
    paint(pos, 1.0, full, geom, resampler='tsc')

    got = numpy.zeros((16, 16, 16))
    pos_soa = dev(pos).t().contiguous()
    for x0 in (0, 8):
        slab = torch.zeros((8, 16, 16), dtype=torch.float64, device='cuda')
        hiplib.check(lib.nbk_paint_f64(
            hiplib.dptr(pos_soa), None, len(pos),
            hiplib.i64_arr(geom.Nmesh), hiplib.f64_arr(geom.BoxSize),
            1, 0.0, hiplib.dptr(slab), x0, 8, None), 'paint')
        got[x0:x0 + 8] = host(slab)
    assert_allclose(got, full, rtol=1e-12, atol=1e-12)


# ---- FFT ----------------------------------------------------------------

@pytest.mark.parametrize('N', [8, 16, 32, 64, 128])
def test_r2c_matches_oracle(lib, N):
    geom = MeshGeometry(N, 100.)
    field = numpy.random.RandomState(7).normal(size=(N, N, N))
    want = r2c(field, geom)

    real_t = dev(field)
    nzh = N // 2 + 1
    cplx_t = torch.empty((N, N, nzh), dtype=torch.complex128,
                         device='cuda')
    scale = 1.0 / N ** 3
    hiplib.check(lib.nbk_fft_r2c_z(
        hiplib.dptr(real_t), hiplib.dptr(cplx_t), N * N, N, scale, None),
        'z')
    hiplib.check(lib.nbk_fft_c_strided(
        hiplib.dptr(cplx_t), N, nzh, N, N * nzh, nzh, -1, None), 'y')
    hiplib.check(lib.nbk_fft_c_strided(
        hiplib.dptr(cplx_t), N, N * nzh, 1, 0, N * nzh, -1, None), 'x')
    got = host(cplx_t)
    assert_allclose(got, want, rtol=1e-11, atol=1e-13)


@pytest.mark.parametrize('N', [8, 32, 64])
def test_c2r_roundtrip(lib, N):
    geom = MeshGeometry(N, 50.)
    field = numpy.random.RandomState(8).normal(size=(N, N, N))
    nzh = N // 2 + 1

    real_t = dev(field)
    cplx_t = torch.empty((N, N, nzh), dtype=torch.complex128,
                         device='cuda')
    hiplib.check(lib.nbk_fft_r2c_z(
        hiplib.dptr(real_t), hiplib.dptr(cplx_t), N * N, N,
        1.0 / N ** 3, None), 'z')
    hiplib.check(lib.nbk_fft_c_strided(
        hiplib.dptr(cplx_t), N, nzh, N, N * nzh, nzh, -1, None), 'y')
    hiplib.check(lib.nbk_fft_c_strided(
        hiplib.dptr(cplx_t), N, N * nzh, 1, 0, N * nzh, -1, None), 'x')

    # inverse
    hiplib.check(lib.nbk_fft_c_strided(
        hiplib.dptr(cplx_t), N, N * nzh, 1, 0, N * nzh, +1, None), 'xi')
    hiplib.check(lib.nbk_fft_c_strided(
        hiplib.dptr(cplx_t), N, nzh, N, N * nzh, nzh, +1, None), 'yi')
    back_t = torch.empty((N, N, N), dtype=torch.float64, device='cuda')
    hiplib.check(lib.nbk_fft_c2r_z(
        hiplib.dptr(cplx_t), hiplib.dptr(back_t), N * N, N, None), 'zi')
    assert_allclose(host(back_t), field, rtol=1e-11, atol=1e-12)


def test_fft_rejects_bad_length(lib):
    t = torch.zeros(24, dtype=torch.float64, device='cuda')
    c = torch.zeros(14, dtype=torch.complex128, device='cuda')
    rc = lib.nbk_fft_r2c_z(hiplib.dptr(t), hiplib.dptr(c), 2, 12, 1.0,
                           None)
    assert rc == -3     # NBK_ERR_UNSUPPORTED


# ---- k-space elementwise ------------------------------------------------

@pytest.mark.parametrize('window', ['cic', 'tsc', 'pcs'])
@pytest.mark.parametrize('interlaced', [False, True])
def test_compensate_matches_oracle(lib, window, interlaced):
    geom = MeshGeometry(16, 32.)
    rng = numpy.random.RandomState(9)
    c = (rng.normal(size=geom.cshape)
         + 1j * rng.normal(size=geom.cshape)).astype('c16')
    want = c.copy()
    apply_compensation(want, geom, window, interlaced)

    c_t = dev(c)
    dims = hiplib.i64_arr(geom.cshape)
    off = hiplib.i64_arr((0, 0, 0))
    hiplib.check(lib.nbk_compensate_f64(
        hiplib.dptr(c_t), hiplib.i64_arr(geom.Nmesh), dims, off, None,
        hiplib.WINDOW_IDS[window], int(interlaced), None), 'comp')
    assert_allclose(host(c_t), want, rtol=1e-12, atol=1e-12)


def test_interlace_combine_matches_oracle(lib):
    geom = MeshGeometry(16, 32.)
    rng = numpy.random.RandomState(10)
    c1 = (rng.normal(size=geom.cshape)
          + 1j * rng.normal(size=geom.cshape)).astype('c16')
    c2 = (rng.normal(size=geom.cshape)
          + 1j * rng.normal(size=geom.cshape)).astype('c16')

    k = complex_circular_coords(geom)    # w = k*H
    kH = k[0] + k[1] + k[2]
    want = c1 * 0.5 + c2 * 0.5 * numpy.exp(0.5j * kH)

    c1_t, c2_t = dev(c1), dev(c2)
    hiplib.check(lib.nbk_interlace_combine_f64(
        hiplib.dptr(c1_t), hiplib.dptr(c2_t),
        hiplib.i64_arr(geom.Nmesh), hiplib.f64_arr(geom.BoxSize),
        hiplib.i64_arr(geom.cshape), hiplib.i64_arr((0, 0, 0)), None,
        None), 'interlace')
    assert_allclose(host(c1_t), want, rtol=1e-12, atol=1e-12)


@pytest.mark.parametrize('poles', [[], [0, 2, 4]])
@pytest.mark.parametrize('los', [(0, 0, 1), (1, 0, 0)])
def test_bin_kernel_matches_oracle(lib, poles, los):
    from oracle import project_to_basis as oracle_bin
    geom = MeshGeometry(16, 32.)
    rng = numpy.random.RandomState(12)
    y3d = (rng.normal(size=geom.cshape)
           + 1j * rng.normal(size=geom.cshape)).astype('c16')

    kedges = numpy.arange(0., numpy.pi * 16 / 32. + 0.3, 0.15)
    Nmu = 4
    muedges = numpy.linspace(-1, 1, Nmu + 1)
    want, want_poles = oracle_bin(y3d, geom, [kedges, muedges], los=los,
                                  poles=poles)

    _poles = sorted(set([0] + list(poles)))
    Nell = len(_poles)
    Nx = len(kedges) - 1
    NB = (Nx + 2) * (Nmu + 2)
    nfields = 3 + 2 * Nell
    sums = torch.zeros(nfields * NB, dtype=torch.float64, device='cuda')
    y_t = dev(y3d)
    k2_t = dev(kedges ** 2)
    mu_t = dev(muedges)
    hiplib.check(lib.nbk_bin_power_f64(
        hiplib.dptr(y_t), hiplib.i64_arr(geom.Nmesh),
        hiplib.f64_arr(geom.BoxSize), hiplib.i64_arr(geom.cshape),
        hiplib.i64_arr((0, 0, 0)), None,
        hiplib.dptr(k2_t), len(kedges), hiplib.dptr(mu_t), len(muedges),
        hiplib.f64_arr(los), hiplib.int_arr(_poles), Nell, 0,
        hiplib.dptr(sums), hiplib.dptr(sums[NB:]),
        hiplib.dptr(sums[2 * NB:]), hiplib.dptr(sums[3 * NB:]), None),
        'bin')
    h = host(sums)
    shape = (Nx + 2, Nmu + 2)
    xsum = h[:NB].reshape(shape)
    musum = h[NB:2 * NB].reshape(shape)
    Nsum = numpy.round(h[2 * NB:3 * NB]).astype('i8').reshape(shape)
    ys = h[3 * NB:].reshape(Nell, 2, NB)
    ysum = (ys[:, 0] + 1j * ys[:, 1]).reshape((Nell,) + shape)

    # fold + normalize like the host tail
    ysum[..., -2] += ysum[..., -1]
    musum[:, -2] += musum[:, -1]
    xsum[:, -2] += xsum[:, -1]
    Nsum[:, -2] += Nsum[:, -1]
    sl = slice(1, -1)
    with numpy.errstate(invalid='ignore', divide='ignore'):
        y2d = (ysum[0] / Nsum)[sl, sl]
        xm = (xsum / Nsum)[sl, sl]
        mum = (musum / Nsum)[sl, sl]
    xmean_2d, mumean_2d, want_y2d, want_N = want
    assert_allclose(Nsum[sl, sl], want_N)
    assert_allclose(numpy.nan_to_num(xm), numpy.nan_to_num(xmean_2d),
                    rtol=1e-12, atol=1e-12)
    assert_allclose(numpy.nan_to_num(mum), numpy.nan_to_num(mumean_2d),
                    rtol=1e-12, atol=1e-12)
    assert_allclose(numpy.nan_to_num(y2d), numpy.nan_to_num(want_y2d),
                    rtol=1e-11, atol=1e-12)
    if poles:
        w_k, w_poles, w_N = want_poles
        N_1d = Nsum[sl, sl].sum(axis=-1)
        pole_arr = ysum[:, sl, sl].sum(axis=-1) / N_1d
        for i, ell in enumerate(poles):
            j = _poles.index(ell)
            assert_allclose(numpy.nan_to_num(pole_arr[j]),
                            numpy.nan_to_num(w_poles[i]),
                            rtol=1e-11, atol=1e-12)


def test_power3d_matches_oracle(lib):
    geom = MeshGeometry(16, 32.)
    rng = numpy.random.RandomState(11)
    c1 = (rng.normal(size=geom.cshape)
          + 1j * rng.normal(size=geom.cshape)).astype('c16')
    c2 = (rng.normal(size=geom.cshape)
          + 1j * rng.normal(size=geom.cshape)).astype('c16')
    want = compute_3d_power(c1, c2, geom)

    out_t = torch.empty(geom.cshape, dtype=torch.complex128, device='cuda')
    c1_t, c2_t = dev(c1), dev(c2)    # keep alive past the async launch
    hiplib.check(lib.nbk_power3d_f64(
        hiplib.dptr(out_t), hiplib.dptr(c1_t), hiplib.dptr(c2_t),
        float(numpy.prod(geom.BoxSize)), hiplib.i64_arr(geom.cshape),
        hiplib.i64_arr((0, 0, 0)), 1, None), 'power3d')
    assert_allclose(host(out_t), want, rtol=1e-12, atol=1e-12)


# ---- partitioned layouts (the multi-GPU slab/pencil paths) -------------

def test_bin_partitioned_matches_full(lib):
    """binning two y-chunks with off=(0,y0,0) and summing == binning the
    full field (the N>1 layout at fftpower.py:669-672)"""
    geom = MeshGeometry(16, 32.)
    rng = numpy.random.RandomState(13)
    y3d = (rng.normal(size=geom.cshape)
           + 1j * rng.normal(size=geom.cshape)).astype('c16')
    kedges = numpy.arange(0., numpy.pi * 16 / 32. + 0.3, 0.15)
    muedges = numpy.linspace(-1, 1, 4)
    Nx = len(kedges) - 1
    NB = (Nx + 2) * (len(muedges) - 1 + 2)
    nf = 3 + 2

    def run_bin(block, off):
        sums = torch.zeros(nf * NB, dtype=torch.float64, device='cuda')
        b_t = dev(block)
        k2_t = dev(kedges ** 2)
        mu_t = dev(muedges)
        hiplib.check(lib.nbk_bin_power_f64(
            hiplib.dptr(b_t), hiplib.i64_arr(geom.Nmesh),
            hiplib.f64_arr(geom.BoxSize),
            hiplib.i64_arr(block.shape), hiplib.i64_arr(off), None,
            hiplib.dptr(k2_t), len(kedges), hiplib.dptr(mu_t),
            len(muedges), hiplib.f64_arr((0, 0, 1)),
            hiplib.int_arr([0]), 1, 0,
            hiplib.dptr(sums), hiplib.dptr(sums[NB:]),
            hiplib.dptr(sums[2 * NB:]), hiplib.dptr(sums[3 * NB:]),
            None), 'bin')
        return host(sums)

    full = run_bin(y3d, (0, 0, 0))
    # partition along x (slab) and along y (transposed pencil layout)
    hx = run_bin(y3d[:8], (0, 0, 0)) + run_bin(y3d[8:], (8, 0, 0))
    hy = run_bin(numpy.ascontiguousarray(y3d[:, :8]), (0, 0, 0)) \
        + run_bin(numpy.ascontiguousarray(y3d[:, 8:]), (0, 8, 0))
    assert_allclose(hx, full, rtol=1e-12, atol=1e-12)
    assert_allclose(hy, full, rtol=1e-12, atol=1e-12)


def test_compensate_partitioned_matches_full(lib):
    geom = MeshGeometry(16, 32.)
    rng = numpy.random.RandomState(14)
    c = (rng.normal(size=geom.cshape)
         + 1j * rng.normal(size=geom.cshape)).astype('c16')

    full_t = dev(c)
    hiplib.check(lib.nbk_compensate_f64(
        hiplib.dptr(full_t), hiplib.i64_arr(geom.Nmesh),
        hiplib.i64_arr(geom.cshape), hiplib.i64_arr((0, 0, 0)), None,
        0, 0, None), 'c')
    want = host(full_t)

    got = numpy.empty_like(c)
    for y0 in (0, 8):
        blk = dev(numpy.ascontiguousarray(c[:, y0:y0 + 8]))
        hiplib.check(lib.nbk_compensate_f64(
            hiplib.dptr(blk), hiplib.i64_arr(geom.Nmesh),
            hiplib.i64_arr(blk.shape), hiplib.i64_arr((0, y0, 0)), None,
            0, 0, None), 'c')
        got[:, y0:y0 + 8] = host(blk)
    assert_allclose(got, want, rtol=1e-14, atol=1e-14)


@pytest.mark.parametrize('window', ['tsc', 'pcs'])
@pytest.mark.parametrize('shift', [0.0, 0.5])
def test_paint_sorted_matches_oracle(lib, window, shift):
    """LDS-windowed sorted-paint path == oracle, incl. the interlacing
    shift and box-edge wraps (input sorted by cell like the bucket
    sort's output)"""
    geom = MeshGeometry(16, 32.)
    pos = rand_positions(20000, 32., seed=31)
    cell = ((pos / 2.0).astype('i8') * [16 * 16, 16, 1]).sum(axis=1)
    pos = pos[numpy.argsort(cell)]
    mass = numpy.random.RandomState(32).uniform(0.5, 1.5, size=len(pos))

    want = numpy.zeros((16, 16, 16))
    paint(pos, mass, want, geom, resampler=window, shift=shift)

    mesh_t = torch.zeros((16, 16, 16), dtype=torch.float64, device='cuda')
    pos_soa = dev(pos).t().contiguous()
    mass_t = dev(mass)
    hiplib.check(lib.nbk_paint_sorted_f64(
        hiplib.dptr(pos_soa), hiplib.dptr(mass_t), len(pos),
        hiplib.i64_arr(geom.Nmesh), hiplib.f64_arr(geom.BoxSize),
        hiplib.WINDOW_IDS[window], shift,
        hiplib.dptr(mesh_t), 0, 16, None), 'paint_sorted')
    assert_allclose(host(mesh_t), want, rtol=1e-12, atol=1e-12)


def test_paint_sorted_slab_bounds(lib):
    geom = MeshGeometry(16, 32.)
    pos = rand_positions(8000, 32., seed=33)
    cell = ((pos / 2.0).astype('i8') * [16 * 16, 16, 1]).sum(axis=1)
    pos = pos[numpy.argsort(cell)]
    full = numpy.zeros((16, 16, 16))
    paint(pos, 1.0, full, geom, resampler='tsc')
    got = numpy.zeros((16, 16, 16))
    pos_soa = dev(pos).t().contiguous()
    for x0 in (0, 8):
        slab = torch.zeros((8, 16, 16), dtype=torch.float64, device='cuda')
        hiplib.check(lib.nbk_paint_sorted_f64(
            hiplib.dptr(pos_soa), None, len(pos),
            hiplib.i64_arr(geom.Nmesh), hiplib.f64_arr(geom.BoxSize),
            1, 0.0, hiplib.dptr(slab), x0, 8, None), 'paint_sorted')
        got[x0:x0 + 8] = host(slab)
    assert_allclose(got, full, rtol=1e-12, atol=1e-12)


# ---------------------------------------------------------------------------
# two-level atomic-free sort + ownership-gather paint (nbk_xsort_* /
# nbk_bucket_fine_f64 / nbk_paint_gather_f64) — production C4 path,
# exercised here by shrinking the size thresholds
# ---------------------------------------------------------------------------

@pytest.mark.parametrize('resampler,interlaced', [
    ('cic', False), ('cic', True), ('tsc', False), ('tsc', True),
    ('pcs', False), ('pcs', True)])
def test_two_level_gather_paint_matches_direct(resampler, interlaced):
    from nbodykit_amd import set_options
    from nbodykit_amd.lab import ArrayCatalog
    rng = numpy.random.RandomState(17)
    n = 200000
    # clumpy + scrambled: uniform plus a dense blob, shuffled
    pos = numpy.concatenate([
        rng.uniform(0, 64., size=(n, 3)),
        rng.normal(32., 1.5, size=(n // 4, 3)) % 64.])
    rng.shuffle(pos)
    cat = ArrayCatalog({'Position': pos})
    kw = dict(Nmesh=64, BoxSize=64., dtype='f8', resampler=resampler,
              interlaced=interlaced, compensated=False)

    r_direct = cat.to_mesh(**kw).compute(mode='real')
    with set_options(sort_min_n=1024, sort_two_level_min_n=1024,
                     sort_two_level_min_cells=1):
        r_two = cat.to_mesh(**kw).compute(mode='real')
    assert_allclose(r_two.value.cpu().numpy(),
                    r_direct.value.cpu().numpy(), rtol=1e-12, atol=1e-12)


def test_two_level_gather_paint_weighted():
    from nbodykit_amd import set_options
    from nbodykit_amd.lab import ArrayCatalog
    rng = numpy.random.RandomState(18)
    n = 150000
    cat = ArrayCatalog({'Position': rng.uniform(0, 32., size=(n, 3)),
                        'Weight': rng.exponential(size=n)})
    kw = dict(Nmesh=32, BoxSize=32., dtype='f8', compensated=False)
    r_direct = cat.to_mesh(**kw).compute(mode='real')
    with set_options(sort_min_n=1024, sort_two_level_min_n=1024,
                     sort_two_level_min_cells=1):
        r_two = cat.to_mesh(**kw).compute(mode='real')
    assert_allclose(r_two.value.cpu().numpy(),
                    r_direct.value.cpu().numpy(), rtol=1e-12, atol=1e-12)


def test_two_level_gather_paint_chunked():
    # chunking accumulates across gather paints (accumulate=1 path)
    from nbodykit_amd import set_options
    from nbodykit_amd.lab import ArrayCatalog
    rng = numpy.random.RandomState(19)
    n = 120000
    cat = ArrayCatalog({'Position': rng.uniform(0, 32., size=(n, 3))})
    kw = dict(Nmesh=32, BoxSize=32., dtype='f8', compensated=False)
    r_direct = cat.to_mesh(**kw).compute(mode='real')
    with set_options(sort_min_n=1024, sort_two_level_min_n=1024,
                     sort_two_level_min_cells=1,
                     paint_chunk_size=33333):
        r_two = cat.to_mesh(**kw).compute(mode='real')
    assert_allclose(r_two.value.cpu().numpy(),
                    r_direct.value.cpu().numpy(), rtol=1e-12, atol=1e-12)


def test_gather_paint_slab_partition():
    """The multi-GPU slab path of the gather kernel: painting each half
    slab separately (x0/nx_local) and stacking must equal the full
    paint.  Ghost handling: each half is given ALL particles (a
    superset of routed local+ghost particles — the kernel must ignore
    out-of-slab tiles by construction)."""
    import torch
    from nbodykit_amd import set_options
    from nbodykit_amd.source.mesh.catalog import _prepare_particles
    from nbodykit_amd.pm import ParticleMesh

    rng = numpy.random.RandomState(23)
    n = 150000
    N = 32
    pos = rng.uniform(0, 64., size=(n, 3))
    pm = ParticleMesh(BoxSize=64., Nmesh=[N, N, N], dtype='f8')
    pos_t = torch.as_tensor(pos).to('cuda')

    with set_options(sort_min_n=1024, sort_two_level_min_n=1024,
                     sort_two_level_min_cells=1):
        soa, mass, sorted_, rowtab = _prepare_particles(pos_t, None, pm)
    assert rowtab is not None
    if isinstance(rowtab, tuple):
        table, pair_gs, n_eff = rowtab
    else:
        table, pair_gs, n_eff = rowtab, -1, n

    lib = hiplib.require()
    nmesh = hiplib.i64_arr(pm.Nmesh)
    box = hiplib.f64_arr(pm.BoxSize)

    full = torch.zeros((N, N, N), dtype=torch.float64, device='cuda')
    hiplib.check(lib.nbk_paint_gather_f64(
        hiplib.dptr(soa), None, n_eff, nmesh, box, 0, 0.0,
        hiplib.dptr(table), hiplib.dptr(full), 0, N, 0, pair_gs,
        hiplib.cur_stream()), 'gather full')

    parts = []
    for x0 in (0, N // 2):
        slab = torch.zeros((N // 2, N, N), dtype=torch.float64,
                           device='cuda')
        hiplib.check(lib.nbk_paint_gather_f64(
            hiplib.dptr(soa), None, n_eff, nmesh, box, 0, 0.0,
            hiplib.dptr(table), hiplib.dptr(slab), x0, N // 2, 0,
            pair_gs, hiplib.cur_stream()), 'gather slab')
        parts.append(slab)
    stacked = torch.cat(parts, dim=0)
    torch.cuda.synchronize()
    assert_allclose(stacked.cpu().numpy(), full.cpu().numpy(),
                    rtol=1e-13, atol=1e-13)
    # sanity: total mass conserved
    assert abs(full.sum().item() - n) < 1e-6
