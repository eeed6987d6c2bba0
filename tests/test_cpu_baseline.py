"""
The bench's 8-core CPU-baseline harness must compute the SAME thing the
oracle does — it is the reference path being timed, not a lookalike.
Checks the slab-decomposed fork-worker paint against the plain oracle
paint (exact), and the end-to-end harness runs on a small config.
"""
import multiprocessing as mp
from multiprocessing import shared_memory

import numpy
import pytest


@pytest.mark.parametrize('resampler', ['cic', 'tsc', 'pcs'])
def test_slab_paint_matches_oracle(resampler):
    import bench
    from oracle.mesh import MeshGeometry
    from oracle.paint import paint

    nmesh, box, cores = 32, 100., 4
    n = 20000
    rng = numpy.random.RandomState(1)
    pos = rng.uniform(0, box, size=(n, 3))
    geom = MeshGeometry(Nmesh=nmesh, BoxSize=box)
    want = numpy.zeros((nmesh,) * 3)
    paint(pos, numpy.ones(n), want, geom, resampler=resampler)

    shm = shared_memory.SharedMemory(create=True, size=nmesh ** 3 * 8)
    try:
        mesh = numpy.ndarray((nmesh,) * 3, dtype='f8', buffer=shm.buf)
        mesh[:] = 0
        nx_l = nmesh // cores
        b0 = numpy.floor(pos[:, 0] / (box / nmesh)).astype('i8')
        dmin, dmax = {'cic': (0, 1), 'tsc': (-1, 2),
                      'pcs': (-1, 2)}[resampler]
        parts = []
        for r in range(cores):
            lo, hi = r * nx_l, (r + 1) * nx_l
            owners = numpy.zeros(n, dtype=bool)
            for d in range(dmin, dmax + 1):
                c = numpy.remainder(b0 + d, nmesh)
                owners |= (c >= lo) & (c < hi)
            parts.append(numpy.flatnonzero(owners))
        bench._BASE.update(nmesh=nmesh, box=box, cores=cores,
                           shm=shm.name, pos=pos, parts=parts,
                           resampler=resampler)
        ctx = mp.get_context('fork')
        with ctx.Pool(cores) as pool:
            pool.map(bench._baseline_paint_worker, range(cores))
        assert numpy.abs(mesh - want).max() == 0.0
    finally:
        bench._BASE.clear()
        shm.close()
        shm.unlink()


@pytest.mark.timeout(300)
def test_cpu_baseline_harness_runs():
    import bench
    cfg = dict(catalog='uniform', particles=int(2e5), nmesh=64,
               box=250., resampler='cic', interlaced=False, mode='1d')
    out = bench.run_cpu_baseline(cfg, cores=2)
    assert out['cores'] == 2
    assert out['value'] > 0
    assert '64^3 mesh' in out['sample']
