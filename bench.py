"""
bench.py — FFTPower throughput on MI355X (the BASELINE.json metric:
"FFTPower wall-clock (s) + particles/s painted, 1e9 pts/1024^3 mesh
@1/2/4/8 GPU").

One step = one full FFTPower through the public API (paint -> R2C FFT ->
compensation -> |delta(k)|^2 -> k-binning) over a GPU-resident catalog
(no PCIe inside the timed region).  Default N=1 workload = C4 on one
GPU: LogNormal-config 1e9 particles, 1024^3 mesh, CIC, compensated,
mode '1d'.  The catalog is generated ON-GPU with the same statistical
config as BASELINE's (LogNormal: our whitenoise -> EH P(k) scaling ->
Zel'dovich -> lognormal Poisson sampling, all through our own FFT
kernels); exact-RNG parity catalogs are exercised by tests/, not here.

Multi-rank: launched by torch.distributed.run with one rank per GPU
(RCCL); total work fixed (BASELINE quotes 1e9 pts at every GPU count)
=> scaling = "strong".

Output: ONE JSON line from rank 0, with `roofline` (paint-kernel
achieved HBM GB/s from HIP events vs the 8 TB/s peak) and
`cpu_baseline` (the numpy oracle timed on this box's host cores on a
bounded sample).
"""
import argparse
import json
import math
import os
import sys
import time

import numpy

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

HBM_PEAK = 8.0e12          # B/s, MI355X spec (MI355X_MICROARCH.md)
# scatter-kernel algorithmic bytes/particle (pos read + support^3
# deposit read-modify-writes) — applies when the atomic scatter path
# paints (small chunks/meshes)
BYTES_PER_PARTICLE = {'cic': 152.0, 'tsc': 456.0, 'pcs': 1048.0}


def pick_tile(nx_local, n1, n2, pad, span):
    """Mirror of the kernel's nbk_pick_tile (csrc/nbk_paint.hip):
    the (P x-planes x RG y-rows) LDS tile.  Narrow-span windows (CIC)
    run the single-plane specialized kernel with max RG (measured
    faster); wide-span windows minimize the particle re-read factor
    ((P+span)/P) * ((RG+span)/RG) within 160 KiB."""
    budget = 20480 // (n2 + pad)
    best = None
    bP = bRG = 1
    P = 1
    while P <= nx_local and P <= budget and nx_local % P == 0:
        if span <= 1 and P > 1:
            break
        RG = 1
        while RG <= n1 and P * RG <= budget and n1 % RG == 0:
            cost = (1.0 / RG) if span <= 1 \
                else (P + span) / P * (RG + span) / RG
            if best is None or cost < best - 1e-12:
                best = cost
                bP, bRG = P, RG
            RG *= 2
        P *= 2
    return bP, bRG


def gather_bpp(resampler, nmesh, n_particles, interlaced=False,
               fused=True):
    """Algorithmic bytes/particle of the ownership-gather paint, under
    the byte model of the sort mode that actually feeds it.

    PAIR-BUCKET mode (CIC non-interlaced, the default): each (pair,
    group) bucket — holding the pair's two planes of particles plus the
    ~1/RG y-stencil duplicates — is scanned by the 3 tiles whose
    stencils reach the pair, so every 24 B position row is read 3 x
    (1 + span/RG) times, plus the single mesh/spectrum write.  (The
    scheme trades these L2-friendly re-reads for the eliminated 56 GB
    per-row fine sort; DESIGN.md has the minimum-traffic model
    alongside.)

    Row-table mode (TSC/PCS/interlaced): read once per (P-plane x
    RG-row) tile whose stencil reaches it — re-read factor
    ((P+span)/P)*((RG+span)/RG)."""
    sh = bool(interlaced)
    # xhi - xlo per window/shift (csrc/nbk_paint.hip source-span logic)
    span = {('cic', False): 1, ('cic', True): 2,
            ('tsc', False): 3, ('tsc', True): 2,
            ('pcs', False): 3, ('pcs', True): 4}[(resampler, sh)]
    import os
    pair = (resampler == 'cic' and not sh
            and os.environ.get('NBK_SORT_PAIR', '1') != '0')
    if pair:
        from nbodykit_amd.source.mesh.catalog import _pair_gs
        gs = _pair_gs([nmesh, nmesh, nmesh])
        pair = gs is not None
        if pair:
            RG = 1 << gs
            reads = 24.0 * 3.0 * (1.0 + span / float(RG))
    if not pair:
        P, RG = pick_tile(nmesh, nmesh, nmesh, 4 if fused else 0, span)
        reads = 24.0 * (P + span) / P * (RG + span) / RG
    mesh_bytes = 8.0 * nmesh ** 3 / float(n_particles)
    return reads + mesh_bytes


def paint_is_gather(nmesh, n_local):
    """Mirrors the driver's two-level-sort thresholds
    (source/mesh/catalog.py _prepare_particles)."""
    from nbodykit_amd import _global_options as go
    return (nmesh ** 3 > go['sort_two_level_min_cells']
            and n_local >= go['sort_two_level_min_n']
            and nmesh <= 20480)

# (workload, gather?) -> measured HBM bytes per benched paint launch,
# from rocprofv3 --pmc on THIS round's kernels: FETCH_SIZE doubled per
# the gfx950 wide-read half-count + WRITE_SIZE from its own pass
# (profiles/r02_pair_pmc.txt documents the collection).  C4 fused
# paint, pair-sort build: x2-corrected FETCH 30.4*2 = 60.8 + WRITE 8.6
# per launch.  The corrected fetch sits BELOW the 76.5 GB algorithmic
# 3-tile read model because L2 serves part of the shared-bucket
# re-reads.  Refresh whenever the paint kernel changes.
PMC_TRAFFIC_BYTES = {('c4', True): 69.4e9}

WORKLOADS = {
    # BASELINE.json configs (C1 is the CPU-oracle plumbing config)
    'c2': dict(catalog='uniform', particles=int(1e7), nmesh=256,
               box=1000., resampler='cic', interlaced=False, mode='1d'),
    'c3': dict(catalog='lognormal', particles=int(1e8), nmesh=512,
               box=2500., resampler='tsc', interlaced=True, mode='1d'),
    'c4': dict(catalog='lognormal', particles=int(1e9), nmesh=1024,
               box=5000., resampler='cic', interlaced=False, mode='1d'),
    'c5': dict(catalog='lognormal', particles=int(1e9), nmesh=1024,
               box=5000., resampler='cic', interlaced=False, mode='2d',
               cross=True, Nmu=5),
}


def log(rank, *args):
    if rank == 0:
        print('[bench]', *args, file=sys.stderr, flush=True)


# ---- GPU-side 3D FFT helpers for the generator -------------------------

def gpu_r2c(real_t, lib, hiplib):
    import torch
    n0, n1, n2 = real_t.shape
    nzh = n2 // 2 + 1
    out = torch.empty((n0, n1, nzh), dtype=torch.complex128, device='cuda')
    s = hiplib.cur_stream()
    hiplib.check(lib.nbk_fft_r2c_z(hiplib.dptr(real_t), hiplib.dptr(out),
                                   n0 * n1, n2, 1.0 / (n0 * n1 * n2), s),
                 'r2c_z')
    hiplib.check(lib.nbk_fft_c_strided(hiplib.dptr(out), n1, nzh, n0,
                                       n1 * nzh, nzh, -1, s), 'y')
    hiplib.check(lib.nbk_fft_c_strided(hiplib.dptr(out), n0, n1 * nzh, 1,
                                       0, n1 * nzh, -1, s), 'x')
    return out


def gpu_c2r(cplx_t, n2, lib, hiplib):
    import torch
    n0, n1, nzh = cplx_t.shape
    s = hiplib.cur_stream()
    work = cplx_t.clone()
    hiplib.check(lib.nbk_fft_c_strided(hiplib.dptr(work), n0, n1 * nzh, 1,
                                       0, n1 * nzh, +1, s), 'xi')
    hiplib.check(lib.nbk_fft_c_strided(hiplib.dptr(work), n1, nzh, n0,
                                       n1 * nzh, nzh, +1, s), 'yi')
    out = torch.empty((n0, n1, n2), dtype=torch.float64, device='cuda')
    hiplib.check(lib.nbk_fft_c2r_z(hiplib.dptr(work), hiplib.dptr(out),
                                   n0 * n1, n2, s), 'zi')
    return out


# ---- GPU catalog generation --------------------------------------------

def gen_uniform(n_total, box, rank, ws, seed):
    """per-rank slab-local uniform positions (n_total/ws each)"""
    import torch
    n_local = n_total // ws
    g = torch.Generator(device='cuda')
    g.manual_seed(seed * 1000 + rank)
    pos = torch.rand((n_local, 3), generator=g, dtype=torch.float64,
                     device='cuda')
    pos[:, 0] = (pos[:, 0] + rank) * (box / ws)   # rank's x-slab
    pos[:, 1] *= box
    pos[:, 2] *= box
    return pos


def gen_lognormal(n_total, nmesh, box, rank, ws, seed, bias=2.0,
                  redshift=0.55):
    """GPU LogNormal pipeline (same statistical config as the CPU
    generator in nbodykit_amd.mockmaker; torch RNG)."""
    import torch
    from nbodykit_amd import hiplib
    from nbodykit_amd.cosmology import Planck15, LinearPower
    lib = hiplib.require()

    N = int(nmesh)
    L = float(box)
    V = L ** 3
    Ntot = float(N) ** 3
    nbar = n_total / V
    H = L / N

    # deterministic whitenoise, identical on all ranks
    g = torch.Generator(device='cuda')
    g.manual_seed(seed)
    noise = torch.randn((N, N, N), generator=g, dtype=torch.float64,
                        device='cuda')
    delta_k = gpu_r2c(noise, lib, hiplib)
    del noise
    delta_k *= math.sqrt(Ntot)

    # P(k) via a log-k interpolation table (EH, Planck15)
    Plin = LinearPower(Planck15, redshift=redshift,
                       transfer='EisensteinHu')
    ktab = numpy.logspace(-6, numpy.log10(2 * math.pi / L * N * 2), 4096)
    ptab = Plin(ktab)
    lktab = torch.as_tensor(numpy.log(ktab)).to('cuda')
    lptab = torch.as_tensor(numpy.log(ptab)).to('cuda')

    def freqs(n, half=False):
        if half:
            f = torch.arange(n // 2 + 1, dtype=torch.float64,
                             device='cuda')
            f[-1] = -(n // 2)
        else:
            f = torch.arange(n, dtype=torch.float64, device='cuda')
            f[f >= n // 2] -= n
            if n % 2 == 0:
                f[n // 2] = -(n // 2)
        return f * (2 * math.pi / L)

    kx = freqs(N).view(-1, 1, 1)
    ky = freqs(N).view(1, -1, 1)
    kz = freqs(N, half=True).view(1, 1, -1)
    k2 = kx * kx + ky * ky + kz * kz
    kmag = torch.sqrt(k2)
    lk = torch.log(torch.clamp(kmag, min=ktab[0]))
    # linear interpolation in log-log
    idx = torch.clamp(torch.searchsorted(lktab, lk.reshape(-1)), 1,
                      len(ktab) - 1)
    lk0 = lktab[idx - 1]
    lk1 = lktab[idx]
    w = (lk.reshape(-1) - lk0) / (lk1 - lk0)
    lp = lptab[idx - 1] * (1 - w) + lptab[idx] * w
    P = torch.exp(lp).reshape(k2.shape)
    del lk, idx, lk0, lk1, w, lp

    delta_k *= torch.sqrt(P / V)
    delta_k[0, 0, 0] = 0
    del P

    delta = gpu_c2r(delta_k, N, lib, hiplib)

    # lognormal transform with lagrangian bias (mockmaker.py:284-287)
    field = torch.exp((bias - 1.0) * delta)
    del delta
    field /= field.mean()
    cellmean = field * (nbar * H ** 3)
    del field

    # this rank's x-slab
    nx_l = N // ws
    x0 = nx_l * rank
    lam = cellmean[x0:x0 + nx_l].reshape(-1)
    del cellmean
    counts = torch.poisson(lam, generator=g).long()
    del lam

    npart = int(counts.sum().item())
    # cell corner coordinates of each particle (repeat_interleave)
    cell = torch.repeat_interleave(
        torch.arange(nx_l * N * N, device='cuda'), counts)
    del counts
    iz = cell % N
    iy = (cell // N) % N
    ix = cell // (N * N) + x0

    pos = torch.empty((npart, 3), dtype=torch.float64, device='cuda')
    pos[:, 0] = ix.double() * H
    pos[:, 1] = iy.double() * H
    pos[:, 2] = iz.double() * H

    # Zel'dovich displacement per cell, one axis at a time
    k2c = k2.clone()
    k2c[0, 0, 0] = 1.0
    for i, ki in enumerate((kx, ky, kz)):
        disp_k = delta_k * (1j * ki / k2c)
        disp_k[0, 0, 0] = 0
        disp = gpu_c2r(disp_k, N, lib, hiplib)
        del disp_k
        pos[:, i] += disp[x0:x0 + nx_l].reshape(-1)[cell - x0 * N * N]
        del disp
    del delta_k, k2, k2c, cell, ix, iy, iz

    # in-cell uniform shift then periodic wrap
    pos += torch.rand((npart, 3), generator=g, dtype=torch.float64,
                      device='cuda') * H
    pos %= L
    return pos


# ---- main ---------------------------------------------------------------

def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--gpus', type=int, default=1)
    # default K large enough that the timed region dominates the GPU
    # activity of the run (r01 verdict item: sparse SMI sampling can
    # miss a too-short timed window)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument('--warmup', type=int, default=2)
    ap.add_argument('--workload', default='c4',
                    choices=sorted(WORKLOADS))
    ap.add_argument('--particles', type=int, default=None)
    ap.add_argument('--nmesh', type=int, default=None)
    ap.add_argument('--no-cpu-baseline', action='store_true')
    ap.add_argument('--traffic-bytes', type=float, default=None,
                    help='measured HBM bytes per paint launch from '
                         'rocprofv3 --pmc (see profiles/)')
    args = ap.parse_args()

    import torch
    ws = int(os.environ.get('WORLD_SIZE', '1'))
    rank = int(os.environ.get('RANK', '0'))
    if ws > 1:
        import torch.distributed as dist
        local = int(os.environ.get('LOCAL_RANK', rank))
        # modulo lets an N-rank run share fewer devices (single-GPU
        # smoke testing of the multi-rank path); production launches
        # have one rank per GPU and the modulo is a no-op.  RCCL refuses
        # duplicate devices, so shared-device smoke runs fall back to
        # gloo collectives with host staging (pm.all_to_all_tensor) —
        # compute stays on the GPU; such runs check correctness, not
        # interconnect throughput.
        ndev = max(1, torch.cuda.device_count())
        torch.cuda.set_device(local % ndev)
        dist.init_process_group('nccl' if ndev >= ws else 'gloo')

    from nbodykit_amd import profiling, set_options
    from nbodykit_amd.lab import FFTPower
    from nbodykit_amd.source.catalog.device import DeviceArrayCatalog
    from nbodykit_amd.comm import default_comm

    comm = default_comm()
    assert comm.size == ws

    cfg = dict(WORKLOADS[args.workload])
    if args.particles:
        cfg['particles'] = args.particles
    if args.nmesh:
        cfg['nmesh'] = args.nmesh

    n_total = cfg['particles']
    nmesh = cfg['nmesh']
    box = cfg['box']
    cross = cfg.get('cross', False)

    log(rank, 'workload %s: %s %.0e particles, %d^3 mesh, %s%s, %s'
        % (args.workload, cfg['catalog'], n_total, nmesh, cfg['resampler'],
           ' interlaced' if cfg['interlaced'] else '', cfg['mode']))

    t_gen = time.time()
    if cfg['catalog'] == 'uniform':
        pos = gen_uniform(n_total, box, rank, ws, seed=42)
        pos2 = gen_uniform(n_total, box, rank, ws, seed=43) if cross \
            else None
    else:
        half = n_total // 2 if cross else n_total
        pos = gen_lognormal(half, nmesh, box, rank, ws, seed=42)
        pos2 = gen_lognormal(half, nmesh, box, rank, ws, seed=43) \
            if cross else None
    torch.cuda.synchronize()
    log(rank, 'generation done in %.1fs; local particles: %d'
        % (time.time() - t_gen, len(pos)))

    boxarr = numpy.array([box] * 3)
    cat = DeviceArrayCatalog({'Position': pos}, BoxSize=boxarr)
    cat2 = DeviceArrayCatalog({'Position': pos2}, BoxSize=boxarr) \
        if cross else None

    def make_mesh(c):
        return c.to_mesh(Nmesh=nmesh, dtype='f8', compensated=True,
                         resampler=cfg['resampler'],
                         interlaced=cfg['interlaced'])

    fft_kw = dict(mode=cfg['mode'])
    if cfg['mode'] == '2d':
        fft_kw['Nmu'] = cfg.get('Nmu', 5)

    def step():
        r = FFTPower(make_mesh(cat), second=(make_mesh(cat2) if cross
                                             else None), **fft_kw)
        return r

    # one huge chunk: data is GPU-resident, the chunk loop is vestigial
    with set_options(paint_chunk_size=1 << 30):
        for _ in range(args.warmup):
            step()
        torch.cuda.synchronize()
        comm.barrier()

        profiling.enable()
        t0 = time.time()
        for _ in range(args.steps):
            result = step()
        torch.cuda.synchronize()
        comm.barrier()
        elapsed = time.time() - t0
    elapsed = comm.allreduce(elapsed, op='max')

    prof = profiling.summary()

    # one untimed pure-paint launch (real-mesh gather, no fused z-FFT)
    # so the bench line can quote BOTH fractions (r01 verdict item 6):
    # the fused kernel contains FFT work its byte model doesn't count
    pure_paint = None
    gather0 = paint_is_gather(nmesh, len(pos))
    if rank == 0 and ws == 1 and gather0:
        profiling.reset()
        with set_options(paint_chunk_size=1 << 30):
            # to_real_field directly: compute('real') on a compensated
            # mesh applies the k-space action and would route through
            # the FUSED complex path instead of the pure gather paint
            make_mesh(cat).to_real_field()
        pp = profiling.summary().get('paint')
        if pp and pp['ms'] > 0:
            bpp0 = gather_bpp(cfg['resampler'], nmesh,
                              max(1, len(pos)),
                              interlaced=cfg['interlaced'], fused=False)
            ach0 = pp['units'] * bpp0 / (pp['ms'] * 1e-3)
            pure_paint = {
                'kernel': 'nbk_paint_gather_f64[%s]' % cfg['resampler'],
                'ms_per_launch': pp['ms'] / pp['calls'],
                'algorithmic_B_per_particle': bpp0,
                'achieved_GBps': ach0 / 1e9,
                'frac': ach0 / HBM_PEAK,
            }
    profiling.disable()

    n_global = comm.allreduce(len(pos)) + (comm.allreduce(len(pos2))
                                           if cross else 0)
    value = n_global * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1e3

    # paint-kernel roofline (rank-0 local figures)
    paint = prof.get('paint', {'ms': 0.0, 'calls': 0, 'units': 0})
    gather = paint_is_gather(nmesh, len(pos))
    if gather:
        bpp = gather_bpp(cfg['resampler'], nmesh, max(1, len(pos)),
                         interlaced=cfg['interlaced'])
    else:
        bpp = BYTES_PER_PARTICLE[cfg['resampler']]
    algo_bytes = paint['units'] * bpp
    achieved = algo_bytes / (paint['ms'] * 1e-3) if paint['ms'] > 0 else 0.
    # measured HBM bytes per paint launch from rocprofv3 --pmc
    # (FETCH_SIZE doubled per the gfx950 wide coalesced-read half-count,
    # plus WRITE_SIZE, collected in separate passes — see
    # profiles/r02_pmc_summary.txt for the runs these come from);
    # --traffic-bytes overrides for fresh measurements
    traffic = args.traffic_bytes
    if traffic is None:
        traffic = PMC_TRAFFIC_BYTES.get((args.workload, gather))

    cpu_baseline = None
    if rank == 0 and ws == 1 and not args.no_cpu_baseline:
        cpu_baseline = run_cpu_baseline(cfg)

    if rank == 0:
        out = {
            'metric': 'FFTPower_particles_per_s',
            'value': value,
            'unit': 'particles/s',
            'n_gpus': ws,
            'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': ms_per_step,
            'higher_is_better': True,
            'scaling': 'strong',
            'vs_baseline': None,
            'dtype': 'f64',
            'data': 'synthetic (GPU-generated %s, BASELINE %s config, '
                    'torch RNG)' % (cfg['catalog'], args.workload.upper()),
            'config': {
                'workload': args.workload.upper(),
                'particles': n_global,
                'nmesh': nmesh,
                'box': box,
                'resampler': cfg['resampler'],
                'interlaced': cfg['interlaced'],
                'mode': cfg['mode'],
                'compensated': True,
                'parallelism': 'slab-dp%d' % ws,
            },
            'roofline': {
                'bound': 'hbm',
                'achieved': achieved / 1e9,
                'peak': HBM_PEAK / 1e9,
                'unit': 'GB/s',
                'frac': achieved / HBM_PEAK,
                'traffic': traffic,
                'kernel': ('nbk_paint_gather_fft_f64[%s]' if gather
                           else 'nbk_paint_f64[%s]') % cfg['resampler'],
                'paint_ms_per_launch': (paint['ms'] / paint['calls']
                                        if paint['calls'] else None),
                'algorithmic_B_per_particle': bpp,
                'pure_paint': pure_paint,
                # mesh-cells/s FFT'd through the STANDALONE strided
                # pass (HIP events; units = Nmesh^3 per pass).  In the
                # deferred path that is the y pass only: the z pass
                # runs fused inside the paint kernel and the x pass
                # inside the binning kernel (nbk_fft_x_bin_f64), so
                # neither appears here.
                'fft_cells_per_s': (
                    prof['fft_strided']['units']
                    / (prof['fft_strided']['ms'] * 1e-3)
                    if prof.get('fft_strided', {}).get('ms') else None),
            },
            'cpu_baseline': cpu_baseline,
            # diagnostic: HBM high-water mark on rank 0 (288 GB/GPU)
            'peak_hbm_gb': round(
                torch.cuda.max_memory_allocated() / 1e9, 2),
        }
        print(json.dumps(out), flush=True)

    if ws > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


# ---- 8-core CPU baseline at the REAL mesh config -----------------------
# The oracle (the restated reference algorithm) slab-decomposed across
# fork workers, mirroring the reference's MPI design: paint into a
# shared-memory mesh with disjoint x-slab ownership, one full-size
# rfftn (scipy, workers=CORES), compensation + |c|^2 V, then
# project_to_basis partial sums per x-range.  Paint is timed on a
# bounded particle sample and scaled linearly in N (paint is
# embarrassingly linear per particle; FFT/compensation/power/binning
# are measured AT THE FULL mesh, never extrapolated).

_BASE = {}          # fork-inherited (copy-on-write) worker inputs


def _baseline_paint_worker(r):
    import numpy as np
    from multiprocessing import shared_memory
    from oracle.paint import (_cic_offsets_weights,
                              _centered_offsets_weights, _tsc_w, _pcs_w)
    nmesh = _BASE['nmesh']
    cores = _BASE['cores']
    nx_l = nmesh // cores
    lo, hi = r * nx_l, (r + 1) * nx_l
    shm = shared_memory.SharedMemory(name=_BASE['shm'])
    mesh = numpy.ndarray((nmesh, nmesh, nmesh), dtype='f8',
                         buffer=shm.buf)
    slab = mesh[lo:hi]
    H = _BASE['box'] / nmesh
    pos = _BASE['pos'][_BASE['parts'][r]]
    resampler = _BASE['resampler']
    CH = 1 << 22
    for s in range(0, len(pos), CH):
        u = pos[s:s + CH] / H
        if resampler == 'cic':
            gen = _cic_offsets_weights(u)
        elif resampler == 'tsc':
            gen = _centered_offsets_weights(u, (-1, 0, 1), _tsc_w, True)
        else:
            gen = _centered_offsets_weights(u, (-1, 0, 1, 2), _pcs_w,
                                            False)
        for ix, iy, iz, w in gen:
            ix = np.remainder(ix.astype('i8'), nmesh)
            sel = (ix >= lo) & (ix < hi)      # own only this slab's rows
            iy = np.remainder(iy.astype('i8')[sel], nmesh)
            iz = np.remainder(iz.astype('i8')[sel], nmesh)
            np.add.at(slab, (ix[sel] - lo, iy, iz), w[sel])
    shm.close()
    return r


def _baseline_bin_worker(r):
    import numpy as np
    from oracle.fftpower import project_to_basis
    lo, hi = _BASE['bin_ranges'][r]
    coords = _BASE['coords']
    sub = [coords[0][lo:hi], coords[1], coords[2]]
    return project_to_basis(_BASE['p3d'][lo:hi], _BASE['geom'],
                            _BASE['edges'], los=(0, 0, 1), coords=sub,
                            _return_sums=True)


def run_cpu_baseline(cfg, cores=8):
    """Reference CPU path (restated oracle) at the benched mesh size on
    `cores` host cores; returns particles/s for the full workload with
    only the paint linearly rescaled from a bounded particle sample."""
    import multiprocessing as mp
    from multiprocessing import shared_memory
    import scipy.fft
    from oracle.mesh import MeshGeometry, complex_coords
    from oracle.fftpower import (apply_compensation, compute_3d_power,
                                 project_to_basis)

    nmesh = cfg['nmesh']
    box = float(cfg['box'])
    resampler = cfg['resampler']
    n_full = cfg['particles']
    # bounded paint sample: ~nmesh^3/8 particles (>= 1e7), never more
    # than the real count
    n_s = int(min(n_full, max(nmesh ** 3 // 8, int(1e7))))
    scale_n = n_full / n_s

    rng = numpy.random.RandomState(42)
    ctx = mp.get_context('fork')

    geom = MeshGeometry(Nmesh=nmesh, BoxSize=box)
    shm = shared_memory.SharedMemory(create=True,
                                     size=nmesh ** 3 * 8)
    try:
        mesh = numpy.ndarray((nmesh, nmesh, nmesh), dtype='f8',
                             buffer=shm.buf)
        mesh[:] = 0.0

        pos = rng.uniform(0, box, size=(n_s, 3))

        # partition by deposit x-range (ghost overlap from the window
        # support, as the reference's decompose does)
        nx_l = nmesh // cores
        b0 = numpy.floor(pos[:, 0] / (box / nmesh)).astype('i8')
        # deposit-cell range relative to floor(x/H): TSC's base is
        # floor(u+0.5)-1, so its cells span [floor(u)-1, floor(u)+2]
        dmin, dmax = {'cic': (0, 1), 'tsc': (-1, 2),
                      'pcs': (-1, 2)}[resampler]
        parts = []
        for r in range(cores):
            lo, hi = r * nx_l, (r + 1) * nx_l
            owners = numpy.zeros(n_s, dtype=bool)
            for d in range(dmin, dmax + 1):
                c = numpy.remainder(b0 + d, nmesh)
                owners |= (c >= lo) & (c < hi)
            parts.append(numpy.flatnonzero(owners))

        _BASE.update(nmesh=nmesh, box=box, cores=cores, shm=shm.name,
                     pos=pos, parts=parts, resampler=resampler)

        t0 = time.time()
        with ctx.Pool(cores) as pool:
            pool.map(_baseline_paint_worker, range(cores))
        t_paint = time.time() - t0

        t0 = time.time()
        nbar = n_s / float(nmesh) ** 3
        mesh /= nbar                      # 1 + delta normalization
        p3d = scipy.fft.rfftn(mesh, workers=cores) / float(nmesh) ** 3
        t_fft = time.time() - t0

        t0 = time.time()
        apply_compensation(p3d, geom, resampler, cfg['interlaced'])
        p3d = compute_3d_power(p3d, p3d, geom)
        t_pow = time.time() - t0

        dk = 2 * numpy.pi / box
        kmax = numpy.pi * nmesh / box + dk / 2
        kedges = numpy.arange(0., kmax, dk)
        Nmu = 1 if cfg['mode'] == '1d' else cfg.get('Nmu', 5)
        muedges = numpy.linspace(-1, 1, Nmu + 1, endpoint=True)
        coords = complex_coords(geom)
        step = (nmesh + cores - 1) // cores
        _BASE.update(p3d=p3d, geom=geom, coords=coords,
                     edges=[kedges, muedges],
                     bin_ranges=[(r * step, min(nmesh, (r + 1) * step))
                                 for r in range(cores)])
        t0 = time.time()
        with ctx.Pool(cores) as pool:
            partials = pool.map(_baseline_bin_worker, range(cores))
        sums = [sum(p[i] for p in partials) for i in range(4)]
        for a in sums:                    # mu-fold (tiny)
            a[..., -2] += a[..., -1]
        t_bin = time.time() - t0
    finally:
        _BASE.clear()
        shm.close()
        shm.unlink()

    t_total = scale_n * t_paint + t_fft + t_pow + t_bin
    return {
        'value': n_full / t_total,
        'unit': 'particles/s',
        'cores': cores,
        'kind': 'port',
        'sample': ('oracle FFTPower at the benched config: %d^3 mesh '
                   '(full-size FFT %.1fs + compensate/power %.1fs + '
                   'binning %.1fs measured as-is), paint timed on '
                   '%.1e of %.1e uniform pts (%.1fs, slab-decomposed '
                   'x%d workers) and scaled linearly in N'
                   % (nmesh, t_fft, t_pow, t_bin, n_s, n_full,
                      t_paint, cores)),
    }


if __name__ == '__main__':
    main()
