"""
Multi-rank GPU parity check — run under torch.distributed.run on a GPU
box (the RCCL path executed on real hardware):

    python -m torch.distributed.run --standalone --nnodes=1 \
        --nproc-per-node 2 tests/dist_gpu_check.py

Ranks share the physical GPU via device-modulo when the box has fewer
GPUs than ranks (single-MI355X world-2 smoke of the multi-rank path);
production launches have one rank per GPU and the modulo is a no-op.

Checks (every rank holds a distinct slice of one shared catalog; the
oracle runs on the full catalog):
  1. fused paint+z-FFT path at ws>1 (CIC compensated, lowered gates) —
     FFTPower P(k) vs oracle at 1e-10
  2. the same with TSC + interlacing (routed ghost range, allgathered
     Hermitian projection, k-space combine on the partitioned layout)
  3. to_real_field at ws>1: each rank's painted 1+delta slab vs the
     oracle mesh slice (exchange/route + gather paint + normalize)
  4. at-scale C2 shape (1e7 uniform pts / 256^3) with DEFAULT gates —
     the real two-level-sort thresholds engaged at ws=2
Writes PASS/FAIL lines to stdout and gpurun_out/dist_check.json.
"""
import json
import os
import sys

import numpy

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

RTOL = 1e-10


def log(*a):
    print('[dist_check]', *a, flush=True)


def rel_err(got, ref):
    got = numpy.ravel(numpy.asarray(got))
    ref = numpy.ravel(numpy.asarray(ref))
    ok = numpy.isfinite(ref) & (numpy.abs(ref) > 0)
    if not ok.any():
        return 0.0
    return float((numpy.abs(got[ok] - ref[ok]) / numpy.abs(ref[ok])).max())


def main():
    import torch
    import torch.distributed as dist
    rank = int(os.environ['RANK'])
    ws = int(os.environ['WORLD_SIZE'])
    local = int(os.environ.get('LOCAL_RANK', rank))
    ndev = max(1, torch.cuda.device_count())
    torch.cuda.set_device(local % ndev)
    # RCCL refuses two ranks on one physical device ("Duplicate GPU
    # detected", RCCL 2.26) — on a box with fewer GPUs than ranks the
    # collectives run over gloo with host staging (pm.all_to_all_tensor)
    # while ALL compute (route/sort/paint/FFT/bin kernels) stays on the
    # GPU; with one rank per GPU the backend is nccl (= RCCL over xGMI).
    backend = 'nccl' if ndev >= ws else 'gloo'
    dist.init_process_group(backend)
    log('rank %d/%d on device %d (of %d), backend %s'
        % (rank, ws, local % ndev, ndev, backend))

    from nbodykit_amd import set_options
    from nbodykit_amd.lab import ArrayCatalog, FFTPower
    from nbodykit_amd.comm import default_comm
    from oracle import fftpower_oracle
    from oracle.catalogmesh import to_real_field as oracle_real
    from oracle.mesh import MeshGeometry

    comm = default_comm()
    assert comm.size == ws

    results = {}
    failures = []

    def shared_uniform(n, box, seed):
        full = numpy.random.RandomState(seed).uniform(0, box, size=(n, 3))
        lo = rank * n // ws
        hi = (rank + 1) * n // ws
        return full, full[lo:hi]

    small_gates = dict(sort_two_level_min_n=1000,
                       sort_two_level_min_cells=100000, sort_min_n=1000)

    # ---- 1. fused multi-rank path, CIC ---------------------------------
    full, mine = shared_uniform(200000, 512., 7)
    cat = ArrayCatalog({'Position': mine}, comm=comm)
    with set_options(**small_gates):
        mesh = cat.to_mesh(Nmesh=128, BoxSize=512., dtype='f8',
                           compensated=True, resampler='cic')
        assert mesh.to_complex_field() is not NotImplemented, \
            'fused path must engage at ws=%d' % ws
        r = FFTPower(mesh, mode='1d', kmin=0.01)
    want = fftpower_oracle(full, Nmesh=128, BoxSize=512., mode='1d',
                           resampler='cic', compensated=True, kmin=0.01)
    e = rel_err(r.power['power'].real, want['power'].real)
    modes_ok = numpy.array_equal(r.power['modes'], want['modes'])
    results['fused_cic'] = e
    log('1. fused CIC ws=%d: rel err %.3g, modes %s' % (ws, e, modes_ok))
    if e > RTOL or not modes_ok:
        failures.append('fused_cic')

    # ---- 2. fused multi-rank path, TSC interlaced ----------------------
    cat2 = ArrayCatalog({'Position': mine}, comm=comm)
    with set_options(**small_gates):
        mesh2 = cat2.to_mesh(Nmesh=128, BoxSize=512., dtype='f8',
                             compensated=True, resampler='tsc',
                             interlaced=True)
        assert mesh2.to_complex_field() is not NotImplemented
        r2 = FFTPower(mesh2, mode='2d', Nmu=5, poles=[0, 2])
    want2 = fftpower_oracle(full, Nmesh=128, BoxSize=512., mode='2d',
                            Nmu=5, poles=[0, 2], resampler='tsc',
                            compensated=True, interlaced=True)
    e2 = rel_err(numpy.nan_to_num(r2.power['power'].real),
                 numpy.nan_to_num(want2['power'].real))
    ep = rel_err(r2.poles['power_0'].real, want2['poles'][0].real)
    results['fused_tsc_interlaced'] = max(e2, ep)
    log('2. fused TSC interlaced ws=%d: rel err %.3g / poles %.3g'
        % (ws, e2, ep))
    if max(e2, ep) > RTOL:
        failures.append('fused_tsc_interlaced')

    # ---- 3. to_real_field slab parity ----------------------------------
    with set_options(**small_gates):
        real = cat.to_mesh(Nmesh=128, BoxSize=512., dtype='f8',
                           resampler='cic').compute(mode='real')
    geom = MeshGeometry(Nmesh=128, BoxSize=512.)
    omesh, oattrs = oracle_real(full, geom, resampler='cic')
    nx_l = 128 // ws
    slab = real.value.cpu().numpy()
    ref_slab = omesh[rank * nx_l:(rank + 1) * nx_l]
    e3 = float(numpy.abs(slab - ref_slab).max() /
               max(1e-300, numpy.abs(ref_slab).max()))
    results['real_slab'] = e3
    log('3. real-field slab ws=%d rank %d: rel err %.3g' % (ws, rank, e3))
    if e3 > RTOL:
        failures.append('real_slab')

    # ---- 4. at-scale C2 shape with DEFAULT gates -----------------------
    full4, mine4 = shared_uniform(int(1e7), 1000., 42)
    cat4 = ArrayCatalog({'Position': mine4}, comm=comm)
    mesh4 = cat4.to_mesh(Nmesh=256, BoxSize=1000., dtype='f8',
                         compensated=True, resampler='cic')
    assert mesh4.to_complex_field() is not NotImplemented, \
        'fused path must engage at default gates for C2 shape'
    r4 = FFTPower(mesh4, mode='1d')
    if rank == 0:
        want4 = fftpower_oracle(full4, Nmesh=256, BoxSize=1000.,
                                mode='1d', resampler='cic',
                                compensated=True)
        e4 = rel_err(r4.power['power'].real, want4['power'].real)
        m4 = numpy.array_equal(r4.power['modes'], want4['modes'])
        results['at_scale_c2'] = e4
        log('4. at-scale C2 ws=%d: rel err %.3g, modes %s' % (ws, e4, m4))
        if e4 > RTOL or not m4:
            failures.append('at_scale_c2')
    comm.barrier()

    # ---- 5. distributed spectral resample (compute(Nmesh=...)) --------
    from oracle import MeshGeometry
    from oracle.mesh import r2c as oracle_r2c
    with set_options(**small_gates):
        m5 = cat.to_mesh(Nmesh=128, BoxSize=512., dtype='f8')
        full5 = m5.compute(mode='real')
        down5 = m5.compute(mode='real', Nmesh=64)
    full_np = numpy.concatenate(
        comm.allgather(full5.value.cpu().numpy()), axis=0)
    down_np = numpy.concatenate(
        comm.allgather(down5.value.cpu().numpy()), axis=0)
    cfull = oracle_r2c(full_np, MeshGeometry(128, 512.))
    cdown = oracle_r2c(down_np, MeshGeometry(64, 512.))
    e5 = float(numpy.abs(cdown[:16, :16, :16]
                         - cfull[:16, :16, :16]).max())
    e5m = abs(down5.cmean() - full5.cmean())
    results['resample'] = max(e5, e5m)
    log('5. distributed resample ws=%d: low-k err %.3g, mean err %.3g'
        % (ws, e5, e5m))
    if max(e5, e5m) > 1e-10:
        failures.append('resample')

    # ---- 6. skewed distribution: every particle in rank 0's slab ------
    # post-routing the other ranks are (nearly) empty — the forced
    # empty-rank row table of the fused path must hold up
    n6 = 120000
    full6 = numpy.random.RandomState(23).uniform(0, 512., size=(n6, 3))
    full6[:, 0] *= (512. / ws) / 512.          # squeeze x into slab 0
    lo = rank * n6 // ws
    hi = (rank + 1) * n6 // ws
    cat6 = ArrayCatalog({'Position': full6[lo:hi]}, comm=comm)
    with set_options(**small_gates):
        mesh6 = cat6.to_mesh(Nmesh=128, BoxSize=512., dtype='f8',
                             compensated=True, resampler='cic')
        assert mesh6.to_complex_field() is not NotImplemented
        r6 = FFTPower(mesh6, mode='1d')
    want6 = fftpower_oracle(full6, Nmesh=128, BoxSize=512., mode='1d',
                            resampler='cic', compensated=True)
    e6 = rel_err(r6.power['power'].real, want6['power'].real)
    m6 = numpy.array_equal(r6.power['modes'], want6['modes'])
    results['skewed'] = e6
    log('6. skewed-slab ws=%d: rel err %.3g, modes %s' % (ws, e6, m6))
    if e6 > RTOL or not m6:
        failures.append('skewed')

    ok = not failures
    # every rank must agree
    all_ok = bool(min(comm.allgather(int(ok))))
    if rank == 0:
        os.makedirs('gpurun_out', exist_ok=True)
        with open('gpurun_out/dist_check.json', 'w') as ff:
            json.dump({'world_size': ws, 'rtol': RTOL, 'pass': all_ok,
                       'failures': failures, 'max_rel_err': results}, ff,
                      indent=1)
        log('RESULT:', 'PASS' if all_ok else 'FAIL', failures)
    dist.destroy_process_group()
    sys.exit(0 if all_ok else 1)


if __name__ == '__main__':
    main()
