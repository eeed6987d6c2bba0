"""
CatalogMesh — the paint driver (reference
nbodykit/source/mesh/catalog.py:11-417): chunked deposit loop with the
``paint_chunk_size`` global option, multi-rank particle routing to slab
owners (the pmesh decompose/exchange step, :271-284, done here with a
torch alltoall over RCCL), interlacing (:289-296, 340-354), particle
counters N/W/W2, shot noise V*W2/W^2 (:378) and the 1+delta
normalization (:394-398).  The deposit itself is the HIP scatter kernel
``nbk_paint_f64``.

The six compensation filter functions (:419-594) keep their reference
names and signatures; ``ComplexField.apply`` recognizes them by identity
and dispatches to the ``nbk_compensate_f64`` kernel.
"""
import logging
import warnings

import numpy

from nbodykit_amd import _global_options
from nbodykit_amd import hiplib, profiling
from nbodykit_amd.base.mesh import MeshSource
from nbodykit_amd.pm import (RealField, ComplexField,
                             exchange_particle_arrays)


def _to_device_f64(arr):
    """numpy or torch -> contiguous f64 CUDA tensor (no copy when it is
    already one: GPU-resident catalogs feed the kernel directly)."""
    import torch
    if isinstance(arr, torch.Tensor):
        t = arr
    else:
        t = torch.as_tensor(numpy.ascontiguousarray(numpy.asarray(arr)))
    return t.to(device='cuda', dtype=torch.float64).contiguous()


def _to_device_mask(arr):
    import torch
    if isinstance(arr, torch.Tensor):
        return arr.to(device='cuda', dtype=torch.bool)
    return torch.as_tensor(numpy.asarray(arr, dtype=bool)).to('cuda')


def _is_trivial_true(sel):
    """all-True zero-stride Selection default: skip the mask entirely"""
    a = sel if isinstance(sel, numpy.ndarray) else None
    return a is not None and a.ndim == 1 and len(a) > 0 \
        and a.strides[0] == 0 and bool(a[0])





def _pair_enabled():
    import os
    return os.environ.get('NBK_SORT_PAIR', '1') != '0'


def _sort_chunk(n, nbuck):
    """Per-block chunk of the counting sorts: sized for ~512 blocks
    (the count matrix shrinks and per-block bandwidth improves with
    bigger chunks — measured 55.8 -> 53.5 ms/step at C4 going
    256K -> 2M — while the grid must still fill 256 CUs), floored so
    the per-block LDS histogram zero/flush (nbuck ints) stays a small
    fraction of the counting work.  NBK_SORT_CHUNK overrides; also
    lets tests prove the consumers are invariant to the chunking."""
    import os
    e = os.environ.get('NBK_SORT_CHUNK')
    if e:
        try:
            return max(4096, min(int(e), 1 << 22))
        except ValueError:
            pass
    target = -(-int(n) // 512)          # ceil(n / 512 blocks)
    return max(16384, 4 * int(nbuck), min(target, 1 << 21))


def _two_level_ys(pm):
    """The y-group shift of the two-level locality sort, or None when no
    LDS budget fits this mesh (coarse histogram n0*(n1>>ys) ints and fine
    window (1<<ys)*n2 ints both <= 40960).  pm-only, so every rank of a
    communicator computes the same answer — the fused to_complex_field
    path gates on it collectively."""
    n0, n1, n2 = (int(x) for x in pm.Nmesh)
    LDSW = 40960
    for ys in range(0, max(1, n1.bit_length())):
        if (1 << ys) > n1 or n1 % (1 << ys):
            break
        win = (1 << ys) * n2
        if win > LDSW:
            break                       # grows with ys: hopeless beyond
        if win < 1024 or win % 1024:
            continue
        if n0 * (n1 >> ys) <= LDSW:
            return ys
    return None


_PAIR_GHOST = {
    # deposit rows relative to iy = floor(y/H): [Dlo, Dhi] covering the
    # stencil at BOTH interlacing shifts when interlaced (the one sorted
    # array feeds both paints); Dlo = dmin, Dhi = dmax + support - 1
    # from the paint launchers' (dmin, dmax) source-span logic
    # (csrc/nbk_paint.hip).  Only the CIC rows are exercised today (the
    # pair sort is gated to CIC non-interlaced); the others are kept
    # correct for any future widening of that gate.
    ('cic', False): (0, 1), ('cic', True): (0, 2),
    ('tsc', False): (-1, 2), ('tsc', True): (-1, 2),
    ('pcs', False): (-1, 2), ('pcs', True): (-1, 3),
}


def _pair_gs(nmesh):
    """The y-group shift of the PAIR-BUCKET duplicating sort, or None
    when the geometry does not admit it (tile = 1 plane x 1<<gs rows x
    (n2+4) doubles in LDS; (n0/2)*(n1>>gs) buckets <= 40960).
    ``nmesh`` is any 3-sequence (pm.Nmesh or a plain triple)."""
    n0, n1, n2 = (int(x) for x in nmesh)
    if n0 % 2:
        return None
    best = None
    for gs in range(1, max(1, n1.bit_length())):
        rg = 1 << gs
        if rg > n1 or n1 % rg:
            break
        if rg * (n2 + 4) * 8 > 160 * 1024:
            break
        if (n0 >> 1) * (n1 >> gs) <= 40960:
            best = gs
    return best


def _prepare_particles(pos_t, mass_t, pm, force_rowtab=False,
                       window='cic', interlaced=False):
    """Return (pos_soa, mass, cell_sorted, table) for the deposit
    kernel, bucket-sorting the chunk by mesh cell when it arrives
    scrambled.  ``table`` is a per-row rowtab tensor (legacy two-level
    sort), a ``(bucket_bases, gs, n_out)`` tuple (pair-bucket
    duplicating sort — the default for big meshes; NBK_SORT_PAIR=0
    reverts), or None (scatter-paint fallback).

    ``force_rowtab`` makes the two-level path run regardless of the size
    thresholds and always emit the row table (the fused paint+z-FFT path
    has committed collectively and must not fall back per-rank: ws>1
    ranks can end up with arbitrarily few local particles after
    routing).  The caller must have checked ``_two_level_ys(pm)`` is
    not None and n2 <= 20480; whether the pair-bucket or the two-level
    row-table pipeline serves the request is decided here from
    (pm, window, interlaced) alone, so the choice is rank-invariant.

    The deposit kernel's wave-merge and its L2 locality both depend on
    nearby-in-space particles being nearby-in-memory; a cell-ordered
    chunk paints ~4x faster than a scrambled one (C4: 206 -> 48 ms
    kernel — the Zel'dovich shift scrambles even generator-ordered
    catalogs across x-planes).  Counting sort (nbk_bucket_count/scatter)
    emits SoA directly; big meshes first run the deterministic chunked
    x-plane pre-sort (nbk_xsort_*, count-matrix based, no global
    atomics) so the cell pass's scattered atomics stay within one
    x-plane's line window.  Cell-ordered inputs skip everything after
    the first count (in-flight order detection).  Part of the timed
    paint path — nothing is cached across calls.
    """
    import torch
    n = len(pos_t)
    n0, n1, n2 = (int(x) for x in pm.Nmesh)
    if force_rowtab and n == 0:
        # empty rank (post-routing): an all-zero table means every
        # range is empty and the gather kernel paints zeros
        gs0 = _pair_gs(pm.Nmesh) if (_pair_enabled() and window == 'cic'
                               and not interlaced) else None
        if gs0 is not None:
            btab = torch.zeros((n0 >> 1) * (n1 >> gs0) + 1,
                               dtype=torch.int32, device='cuda')
            return pos_t.t().contiguous(), mass_t, True, (btab, gs0, 0)
        rowtab = torch.zeros(n0 * n1 + 1, dtype=torch.int32,
                             device='cuda')
        return pos_t.t().contiguous(), mass_t, True, rowtab
    if not force_rowtab and n < _global_options['sort_min_n']:
        return pos_t.t().contiguous(), mass_t, False, None

    lib = hiplib.require()
    ncells = n0 * n1 * n2

    nmesh = hiplib.i64_arr(pm.Nmesh)
    box = hiplib.f64_arr(pm.BoxSize)
    stream = hiplib.cur_stream()

    def count(pos_in, nb, sh, detect):
        counts = torch.zeros(nb, dtype=torch.int32, device='cuda')
        flag = torch.zeros(1, dtype=torch.int32, device='cuda')
        hiplib.check(lib.nbk_bucket_count_f64(
            hiplib.dptr(pos_in), n, nmesh, box, sh, hiplib.dptr(counts),
            hiplib.dptr(flag), stream), 'nbk_bucket_count_f64')
        scrambled = True
        if detect:
            scrambled = int(flag.item()) != 0
        return counts, scrambled

    def scatter(pos_in, m_in, incl, sh, soa):
        # incl = INCLUSIVE int32 bucket cumsum; the kernel's tickets
        # count down to the exclusive base (no host-side shift needed)
        out = torch.empty(3 * n, dtype=torch.float64, device='cuda')
        out_m = None
        if m_in is not None:
            out_m = torch.empty(n, dtype=torch.float64, device='cuda')
        hiplib.check(lib.nbk_bucket_scatter_f64(
            hiplib.dptr(pos_in), hiplib.dptr(m_in), n, nmesh, box, sh,
            int(soa), hiplib.dptr(incl), hiplib.dptr(out),
            hiplib.dptr(out_m), stream), 'nbk_bucket_scatter_f64')
        return out, out_m

    pos_in = pos_t.contiguous()
    # the driver chunks paints at paint_chunk_size (< 2^31), so int32
    # tickets are safe
    assert n < 2 ** 31

    # Big meshes use the two-level ATOMIC-FREE pipeline (the global
    # atomic pipe measures ~25 G ops/s regardless of locality —
    # csrc/count_probe.hip — which bounded the single-level sort):
    # chunked coarse sort by (ix, iy-group) via a per-chunk count
    # matrix + host scan + LDS cursors, then a fused per-bucket fine
    # kernel (LDS count + block scan + placement) emitting the exact
    # cell order.  ys balances the two LDS budgets: coarse histogram
    # n0*(n1>>ys) ints vs fine window (1<<ys)*n2 ints, both <= 40960.
    big = (force_rowtab
           or (n >= _global_options['sort_two_level_min_n']
               and ncells > _global_options['sort_two_level_min_cells']))

    # PAIR-BUCKET duplicating sort: one counting sort by (x-plane pair,
    # y row-group) with stencil-boundary particles duplicated into both
    # touched groups — the per-row fine pass (the single biggest sort
    # cost) disappears, and the gather paint reads whole bucket ranges
    # with its deposit masks dropping the out-of-tile copies.
    # Pair mode pays off when a tile's source planes span a single
    # plane pair boundary (CIC, shift 0: planes [p-1, p] — bucket read
    # by 3 tiles, L2-served).  Wider stencils (TSC/PCS, or CIC's
    # interlaced half-cell shift) span 3+ planes: each bucket gets read
    # by 5-6 tiles and the measured C3 step LOST 2.6 ms — those
    # windows keep the two-level row-table sort.
    pair_ok = (window == 'cic' and not interlaced)
    gs = _pair_gs(pm.Nmesh) if (big and pair_ok and _pair_enabled()
                          and n2 <= 20480) else None
    if gs is not None:
        dlo, dhi = _PAIR_GHOST[(window, bool(interlaced))]
        if dhi - dlo >= (1 << gs):
            gs = None
    if gs is not None:
        nbuck = (n0 >> 1) * (n1 >> gs)
        CH = _sort_chunk(n, nbuck)
        nblocks = (n + CH - 1) // CH
        mat = torch.empty(nblocks * nbuck, dtype=torch.int32,
                          device='cuda')
        hiplib.check(lib.nbk_psort_count_f64(
            hiplib.dptr(pos_in), n, CH, nmesh, box, gs, dlo, dhi,
            hiplib.dptr(mat), stream), 'nbk_psort_count_f64')
        colsum = torch.empty(9 * nbuck, dtype=torch.int32,
                             device='cuda')   # 8 segment partials + sums
        bases = torch.empty(nblocks * nbuck, dtype=torch.int32,
                            device='cuda')
        bucket_bases = torch.empty(nbuck + 1, dtype=torch.int32,
                                   device='cuda')
        hiplib.check(lib.nbk_scan_matrix_i32(
            hiplib.dptr(mat), nblocks, nbuck, hiplib.dptr(colsum),
            hiplib.dptr(bases), hiplib.dptr(bucket_bases), stream),
            'nbk_scan_matrix_i32')
        n_out = int(bucket_bases[-1].item())
        out = torch.empty(3 * n_out, dtype=torch.float64, device='cuda')
        out_m = None
        if mass_t is not None:
            out_m = torch.empty(n_out, dtype=torch.float64,
                                device='cuda')
        hiplib.check(lib.nbk_psort_scatter_f64(
            hiplib.dptr(pos_in), hiplib.dptr(mass_t), n, CH, nmesh, box,
            gs, dlo, dhi, hiplib.dptr(bases), n_out, hiplib.dptr(out),
            hiplib.dptr(out_m), stream), 'nbk_psort_scatter_f64')
        return out, out_m, True, (bucket_bases, gs, n_out)

    ys_fine = _two_level_ys(pm)
    use_two = ys_fine is not None and big
    if use_two:
        ys = ys_fine
        nbuck = n0 * (n1 >> ys)
        CH = _sort_chunk(n, nbuck)
        nblocks = (n + CH - 1) // CH
        mat = torch.empty(nblocks * nbuck, dtype=torch.int32,
                          device='cuda')
        flag = torch.zeros(1, dtype=torch.int32, device='cuda')
        hiplib.check(lib.nbk_xsort_count_f64(
            hiplib.dptr(pos_in), n, CH, nmesh, box, ys, hiplib.dptr(mat),
            hiplib.dptr(flag), stream), 'nbk_xsort_count_f64')
        if not force_rowtab and int(flag.item()) == 0:
            # already cell-ordered: no sorting needed.  (With
            # force_rowtab the sort still runs — it is the cheapest
            # correct way to get the row table the fused path needs.)
            return pos_t.t().contiguous(), mass_t, True, None
        # one fused device scan of the count matrix (replaces a ~4 GB
        # torch transpose/cumsum/sub chain with ~1.5 coalesced passes)
        colsum = torch.empty(9 * nbuck, dtype=torch.int32,
                             device='cuda')   # 8 segment partials + sums
        bases = torch.empty(nblocks * nbuck, dtype=torch.int32,
                            device='cuda')
        bucket_bases = torch.empty(nbuck + 1, dtype=torch.int32,
                                   device='cuda')
        hiplib.check(lib.nbk_scan_matrix_i32(
            hiplib.dptr(mat), nblocks, nbuck, hiplib.dptr(colsum),
            hiplib.dptr(bases), hiplib.dptr(bucket_bases), stream),
            'nbk_scan_matrix_i32')
        coarse = torch.empty(3 * n, dtype=torch.float64, device='cuda')
        mass_c = None
        if mass_t is not None:
            mass_c = torch.empty(n, dtype=torch.float64, device='cuda')
        hiplib.check(lib.nbk_xsort_scatter_f64(
            hiplib.dptr(pos_in), hiplib.dptr(mass_t), n, CH, nmesh, box,
            ys, hiplib.dptr(bases), hiplib.dptr(coarse),
            hiplib.dptr(mass_c), stream), 'nbk_xsort_scatter_f64')
        out = torch.empty(3 * n, dtype=torch.float64, device='cuda')
        out_m = None
        if mass_t is not None:
            out_m = torch.empty(n, dtype=torch.float64, device='cuda')
        # rowtab feeds the ownership-gather paint (tile = RG rows x n2
        # f64 in LDS; needs n2 <= 20480)
        rowtab = None
        if n2 <= 20480:
            rowtab = torch.empty(n0 * n1 + 1, dtype=torch.int32,
                                 device='cuda')
            rowtab[-1] = n
        # with the gather paint downstream only ROW grouping is needed
        # (z order within a row is irrelevant there); the full cell sort
        # serves the nbk_paint_sorted fallback
        rows_only = 1 if rowtab is not None else 0
        hiplib.check(lib.nbk_bucket_fine_f64(
            hiplib.dptr(coarse), hiplib.dptr(mass_c), n, nmesh, box, ys,
            hiplib.dptr(bucket_bases), hiplib.dptr(out),
            hiplib.dptr(out_m), hiplib.dptr(rowtab), rows_only, stream),
            'nbk_bucket_fine_f64')
        return out, out_m, True, rowtab

    counts, scrambled = count(pos_in, ncells, 0, detect=True)
    if not scrambled:
        # already cell-ordered: no scatter needed
        return pos_t.t().contiguous(), mass_t, True, None

    out_soa, out_mass = scatter(
        pos_in, mass_t, torch.cumsum(counts, 0, dtype=torch.int32), 0,
        soa=True)
    return out_soa, out_mass, True, None


def _is_trivial_unit(col):
    """unit-valued zero-stride Weight/Value default: no device copy"""
    a = col if isinstance(col, numpy.ndarray) else None
    return a is not None and a.ndim == 1 and len(a) > 0 \
        and a.strides[0] == 0 and float(a[0]) == 1.0

# union of x-cell offsets a particle can deposit to, relative to
# floor(u0): see DESIGN.md (window support 2/3/4, interlacing shifts by
# +0.5 mesh units)
_GHOST_RANGE = {
    ('cic', False): (0, 1), ('cic', True): (0, 2),
    ('tsc', False): (-1, 2), ('tsc', True): (-1, 2),
    ('pcs', False): (-1, 2), ('pcs', True): (-1, 3),
}


class CatalogMesh(MeshSource):
    logger = logging.getLogger('CatalogMesh')

    def __repr__(self):
        return "(%s as CatalogMesh)" % repr(self.source)

    def __init__(self, source, Nmesh, BoxSize, Position, dtype='f4',
                 resampler='cic', compensated=False, interlaced=False,
                 Value=None, Selection=None, Weight=None, **kwargs):
        from nbodykit_amd.base.catalog import CatalogSourceBase
        assert isinstance(source, CatalogSourceBase)

        self.attrs.update(source.attrs)
        MeshSource.__init__(self, source.comm, Nmesh, BoxSize, dtype)
        self.source = source
        self.dtype = dtype

        self.Position = Position
        self.Weight = Weight
        self.Value = Value
        self.Selection = Selection

        self.attrs['interlaced'] = interlaced
        self.attrs['compensated'] = compensated
        self.attrs['resampler'] = str(resampler)

    # -- properties mirroring the reference (:98-152) ---------------------
    @property
    def interlaced(self):
        return self.attrs['interlaced']

    @interlaced.setter
    def interlaced(self, interlaced):
        self.attrs['interlaced'] = interlaced

    @property
    def resampler(self):
        return self.attrs['resampler']

    @resampler.setter
    def resampler(self, value):
        assert value in ('cic', 'tsc', 'pcs')
        self.attrs['resampler'] = value.lower()

    @property
    def window(self):
        return self.attrs['resampler']

    @window.setter
    def window(self, value):
        self.resampler = value

    @property
    def compensated(self):
        return self.attrs['compensated']

    @compensated.setter
    def compensated(self, value):
        self.attrs['compensated'] = value

    # -- the paint driver -------------------------------------------------
    def to_real_field(self, out=None, normalize=True):
        import torch
        lib = hiplib.require()
        pm = self.pm
        comm = pm.comm

        if out is not None:
            assert isinstance(out, RealField)
            numpy.testing.assert_array_equal(out.pm.Nmesh, pm.Nmesh)
            toret = out
        else:
            toret = RealField(pm)

        interlaced = self.interlaced
        if interlaced:
            real1 = RealField(pm)
            real2 = RealField(pm)

        window_id = hiplib.WINDOW_IDS[self.resampler]
        nmesh = hiplib.i64_arr(pm.Nmesh)
        box = hiplib.f64_arr(pm.BoxSize)
        stream = hiplib.cur_stream()

        Position = self.Position
        Weight = self.Weight
        Value = self.Value
        Selection = self.Selection

        Nlocal = 0
        Wlocal = 0.0
        W2local = 0.0

        # collective chunk loop (reference :303-332); all ranks iterate
        # the same count so the alltoall stays in lockstep
        Nlocalmax = max(comm.allgather(len(Position)))
        chunksize = _global_options['paint_chunk_size']

        i = 0
        while i < max(Nlocalmax, 1):
            s = slice(i, i + chunksize)

            if len(Position) != 0:
                columns = [Position[s]]
                if Weight is not None:
                    columns.append(Weight[s])
                if Value is not None:
                    columns.append(Value[s])
                if Selection is not None:
                    columns.append(Selection[s])
                data = self.source.compute(columns)
                if not isinstance(data, list):
                    data = [data]
                sel = Ellipsis if Selection is None else data.pop()
                if sel is not Ellipsis and _is_trivial_true(sel):
                    sel = Ellipsis
                value = None if Value is None else data.pop()
                weight = None if Weight is None else data.pop()
                position = data.pop()
                if sel is not Ellipsis:
                    sel = _to_device_mask(sel)
                    position = _to_device_f64(position)[sel]
                    if weight is not None:
                        weight = _to_device_f64(weight)[sel]
                    if value is not None:
                        value = _to_device_f64(value)[sel]
            else:
                position = numpy.empty((0, 3), dtype='f8')
                weight = None
                value = None

            # unit-valued broadcast columns (the Weight/Value defaults)
            # never leave the host — the kernel takes mass = NULL
            if weight is not None and _is_trivial_unit(weight):
                weight = None
            if value is not None and _is_trivial_unit(value):
                value = None

            pos_t = _to_device_f64(position)
            w_t = None if weight is None else _to_device_f64(weight)
            v_t = None if value is None else _to_device_f64(value)

            n_chunk = len(pos_t)
            Nlocal += n_chunk
            if w_t is None:
                Wlocal += float(n_chunk)
                W2local += float(n_chunk)
            else:
                Wlocal += float(w_t.sum().item())
                W2local += float((w_t * w_t).sum().item())

            if w_t is None and v_t is None:
                mass_t = None
            elif v_t is None:
                mass_t = w_t
            elif w_t is None:
                mass_t = v_t
            else:
                mass_t = w_t * v_t

            if comm.size > 1:
                pos_t, mass_t = self._route(pos_t, mass_t)

            n = len(pos_t)
            if n > 0:
                pos_soa, mass_t, sorted_, rowtab = _prepare_particles(
                    pos_t, mass_t, pm, window=self.resampler,
                    interlaced=interlaced)
                if isinstance(rowtab, tuple):
                    table, pair_gs, n_eff = rowtab
                else:
                    table, pair_gs, n_eff = rowtab, -1, n
                # fresh fields take plain stores on the first chunk
                acc = 0 if (i == 0 and out is None) else 1
                tgt = (real1 if interlaced else toret).value
                with profiling.collect('paint', n * (1 + interlaced)):
                    if table is not None:
                        hiplib.check(lib.nbk_paint_gather_f64(
                            hiplib.dptr(pos_soa), hiplib.dptr(mass_t),
                            n_eff, nmesh, box, window_id, 0.0,
                            hiplib.dptr(table), hiplib.dptr(tgt),
                            pm.x_start, pm.nx_local, acc, pair_gs,
                            stream), 'nbk_paint_gather_f64')
                        if interlaced:
                            hiplib.check(lib.nbk_paint_gather_f64(
                                hiplib.dptr(pos_soa), hiplib.dptr(mass_t),
                                n_eff, nmesh, box, window_id, 0.5,
                                hiplib.dptr(table),
                                hiplib.dptr(real2.value),
                                pm.x_start, pm.nx_local, acc, pair_gs,
                                stream), 'nbk_paint_gather_f64')
                    else:
                        paint_fn = (lib.nbk_paint_sorted_f64 if sorted_
                                    else lib.nbk_paint_f64)
                        hiplib.check(paint_fn(
                            hiplib.dptr(pos_soa), hiplib.dptr(mass_t), n,
                            nmesh, box, window_id, 0.0,
                            hiplib.dptr(tgt),
                            pm.x_start, pm.nx_local, stream), 'nbk_paint')
                        if interlaced:
                            hiplib.check(paint_fn(
                                hiplib.dptr(pos_soa), hiplib.dptr(mass_t),
                                n, nmesh, box, window_id, 0.5,
                                hiplib.dptr(real2.value),
                                pm.x_start, pm.nx_local, stream),
                                'nbk_paint')
            i = i + chunksize

        if interlaced:
            # k-space combine c = c1/2 + c2/2 exp(i k.H/2) (:341-347)
            c1 = real1.r2c()
            c2 = real2.r2c()
            hiplib.check(lib.nbk_interlace_combine_f64(
                hiplib.dptr(c1.value), hiplib.dptr(c2.value),
                nmesh, box, hiplib.i64_arr(c1.dims), hiplib.i64_arr(c1.off),
                None, stream), 'nbk_interlace_combine_f64')
            combined = c1.c2r()
            toret.value.add_(combined.value)

        N = comm.allreduce(Nlocal)
        W = comm.allreduce(Wlocal)
        W2 = comm.allreduce(W2local)
        nbar = 1.0 * W / float(numpy.prod(pm.Nmesh))

        if N == 0:
            warnings.warn("trying to paint particle source to mesh, but "
                          "no particles were found!", RuntimeWarning)

        with numpy.errstate(divide='ignore', invalid='ignore'):
            shotnoise = float(numpy.prod(pm.BoxSize)) * W2 / W ** 2 \
                if W != 0 else numpy.nan

        toret.attrs = {}
        toret.attrs['shotnoise'] = shotnoise
        toret.attrs['N'] = N
        toret.attrs['W'] = W
        toret.attrs['W2'] = W2
        toret.attrs['num_per_cell'] = nbar

        if normalize:
            if nbar > 0:
                toret.value.div_(nbar)
            else:
                toret.value.fill_(1.0)

        return toret

    def to_complex_field(self, out=None, _defer_x=False):
        """Fused paint -> forward-FFT fast path: the z pass runs inside
        the gather paint's tile flush, so the real mesh never exists in
        HBM; interlaced meshes additionally skip the c2r + re-r2c round
        trip of the real path (combine directly in k).  Numerically the
        FFT is the same radix-2 code as nbk_fft_r2c_z; the 1/N^3 and
        1/nbar (normalize) factors fold into the kernel's output scale.

        ``_defer_x`` (internal, FFTPower's fused x-FFT+binning path;
        only valid for non-interlaced meshes): stop BEFORE the final x
        strided pass and return ``(tensor, n_inner, attrs)`` — the
        pre-x-pass spectrum nbk_fft_x_bin_f64 consumes — instead of a
        ComplexField.  Returns NotImplemented under exactly the same
        (collective) gates as the normal path.

        Multi-rank: every gate below the collective allgather is
        rank-invariant (pm geometry or globally-reduced sizes), so all
        ranks commit to the fused path together — it ends in the RCCL
        pencil transpose (_r2c_finish) and a per-rank fallback would
        deadlock.  Particles are routed to their slab owners first
        (the same ghost exchange as to_real_field) and the two-level
        sort is forced so every rank has a row table even when routing
        leaves it few (or zero) particles.

        Falls back (NotImplemented -> to_real_field().r2c() in
        MeshSource.to_field) for user `out`, paints that need chunking,
        non-power-of-two z, or inputs below the locality-sort
        thresholds."""
        import torch
        if out is not None:
            return NotImplemented
        pm = self.pm
        comm = pm.comm
        ws = comm.size
        n2 = int(pm.Nmesh[2])
        if n2 < 8 or n2 > 4096 or (n2 & (n2 - 1)):
            return NotImplemented
        if _two_level_ys(pm) is None:
            return NotImplemented       # no rowtab possible (pm-based)
        # collective gates: identical decision on every rank.  (The
        # sum()-of-sizes test reduces to the old per-rank test at ws=1.)
        Position = self.Position
        sizes = comm.allgather(len(Position))
        if max(sizes) > _global_options['paint_chunk_size']:
            return NotImplemented
        # cheap gate BEFORE pulling columns: below the two-level sort
        # thresholds the gather/rowtab path cannot engage and the real
        # path would redo all the preparation work (C2-size inputs)
        ncells = int(numpy.prod(pm.Nmesh))
        if not (sum(sizes) >= _global_options['sort_two_level_min_n']
                and ncells > _global_options['sort_two_level_min_cells']):
            return NotImplemented

        lib = hiplib.require()
        interlaced = self.interlaced
        window_id = hiplib.WINDOW_IDS[self.resampler]
        nmesh = hiplib.i64_arr(pm.Nmesh)
        box = hiplib.f64_arr(pm.BoxSize)
        stream = hiplib.cur_stream()

        # single-chunk column pull (the chunk loop of to_real_field)
        sel = None if self.Selection is None else self.Selection
        value = None if self.Value is None else self.Value
        weight = None if self.Weight is None else self.Weight
        data = self.source.compute([c for c in
                                    [Position, weight, value, sel]
                                    if c is not None])
        position = data[0]
        idx = 1
        if weight is not None:
            weight = data[idx]; idx += 1
        if value is not None:
            value = data[idx]; idx += 1
        if sel is not None:
            sel = data[idx]
        if sel is not None and _is_trivial_true(sel):
            sel = None
        if sel is not None:
            m = _to_device_mask(sel)
            position = _to_device_f64(position)[m]
            if weight is not None:
                weight = _to_device_f64(weight)[m]
            if value is not None:
                value = _to_device_f64(value)[m]
        if weight is not None and _is_trivial_unit(weight):
            weight = None
        if value is not None and _is_trivial_unit(value):
            value = None

        pos_t = _to_device_f64(position)
        w_t = None if weight is None else _to_device_f64(weight)
        v_t = None if value is None else _to_device_f64(value)

        Nlocal = len(pos_t)
        if w_t is None:
            Wlocal = float(Nlocal)
            W2local = float(Nlocal)
        else:
            Wlocal = float(w_t.sum().item())
            W2local = float((w_t * w_t).sum().item())
        if w_t is None and v_t is None:
            mass_t = None
        elif v_t is None:
            mass_t = w_t
        elif w_t is None:
            mass_t = v_t
        else:
            mass_t = w_t * v_t

        # global counters (pre-routing, like to_real_field's chunk loop)
        N = comm.allreduce(Nlocal)
        W = comm.allreduce(Wlocal)
        W2 = comm.allreduce(W2local)

        nbar = W / float(numpy.prod(pm.Nmesh))
        if N == 0 or nbar <= 0:
            return NotImplemented       # globally empty: rank-invariant

        if ws > 1:
            pos_t, mass_t = self._route(pos_t, mass_t)

        pos_soa, mass_t, sorted_, rowtab = _prepare_particles(
            pos_t, mass_t, pm, force_rowtab=True,
            window=self.resampler, interlaced=interlaced)
        assert rowtab is not None
        if isinstance(rowtab, tuple):
            table, pair_gs, n_eff = rowtab
        else:
            table, pair_gs, n_eff = rowtab, -1, len(pos_t)
        n_routed = len(pos_t)

        scale = 1.0 / float(numpy.prod(pm.Nmesh)) / nbar
        nzh = n2 // 2 + 1
        shape = (pm.nx_local, int(pm.Nmesh[1]), nzh)

        from nbodykit_amd.pm import _r2c_finish, _r2c_y_transpose

        def one(shift, finish=True):
            z = torch.empty(shape, dtype=torch.complex128, device='cuda')
            # time only the fused kernel as 'paint' (the y/x passes are
            # FFT work, not paint work — the bench roofline reads this)
            with profiling.collect('paint', n_routed):
                hiplib.check(lib.nbk_paint_gather_fft_f64(
                    hiplib.dptr(pos_soa), hiplib.dptr(mass_t), n_eff,
                    nmesh, box, window_id, float(shift),
                    hiplib.dptr(table), hiplib.dptr(z), pm.x_start,
                    pm.nx_local, scale, pair_gs, stream),
                    'nbk_paint_gather_fft_f64')
            if finish:
                return _r2c_finish(z, pm, stream)
            return _r2c_y_transpose(z, pm, stream)

        if _defer_x:
            tensor, n_inner = one(0.0, finish=False)
            tensor2 = None
            if interlaced:
                tensor2, _ = one(0.5, finish=False)
            with numpy.errstate(divide='ignore', invalid='ignore'):
                shotnoise = float(numpy.prod(pm.BoxSize)) * W2 / W ** 2
            return tensor, tensor2, n_inner, {'shotnoise': shotnoise,
                                              'N': N, 'W': W, 'W2': W2,
                                              'num_per_cell': nbar}

        cplx = one(0.0)
        if interlaced:
            c2t = one(0.5)
            c1f = ComplexField(pm, tensor=cplx)
            c2f = ComplexField(pm, tensor=c2t)
            hiplib.check(lib.nbk_interlace_combine_f64(
                hiplib.dptr(c1f.value), hiplib.dptr(c2f.value),
                nmesh, box, hiplib.i64_arr(c1f.dims),
                hiplib.i64_arr(c1f.off), None, stream),
                'nbk_interlace_combine_f64')
            cplx = c1f.value
            # the real path (and the reference, which combines in k
            # then goes through c2r + r2c) Hermitian-projects the
            # self-conjugate z-planes: numpy's irfft drops the
            # imaginary parts of the kz = 0/Nyquist bins, which in k
            # is c <- (c + conj(c(-k)))/2 on those planes.  Multi-rank
            # the field is y-partitioned, and c(-k) needs rows owned by
            # other ranks: allgather the two tiny planes (16 B * Nx * Ny
            # each), project, keep the local y-block.
            self_conj = (0, n2 // 2) if n2 % 2 == 0 else (0,)
            for kz in self_conj:
                if ws > 1:
                    from nbodykit_amd.pm import all_gather_tensor
                    A = cplx[:, :, kz].contiguous()
                    parts = [torch.empty_like(A) for _ in range(ws)]
                    all_gather_tensor(parts, A)
                    Af = torch.cat(parts, dim=1)
                    B = torch.conj(torch.roll(torch.flip(Af, (0, 1)),
                                              (1, 1), (0, 1)))
                    proj = 0.5 * (Af + B)
                    cplx[:, :, kz] = proj[:, pm.y_start:
                                          pm.y_start + pm.ny_local]
                else:
                    A = cplx[:, :, kz]
                    B = torch.conj(torch.roll(torch.flip(A, (0, 1)),
                                              (1, 1), (0, 1)))
                    cplx[:, :, kz] = 0.5 * (A + B)

        with numpy.errstate(divide='ignore', invalid='ignore'):
            shotnoise = float(numpy.prod(pm.BoxSize)) * W2 / W ** 2

        f = ComplexField(pm, tensor=cplx)
        f.attrs = {'shotnoise': shotnoise, 'N': N, 'W': W, 'W2': W2,
                   'num_per_cell': nbar}
        return f

    def _route(self, pos_t, mass_t):
        dmin, dmax = _GHOST_RANGE[(self.resampler, bool(self.interlaced))]
        return route_particles(pos_t, mass_t, self.pm, dmin, dmax)


    @property
    def actions(self):
        actions = MeshSource.actions.fget(self)
        if self.compensated:
            actions = self._get_compensation() + actions
        return actions

    def _get_compensation(self):
        return get_compensation(self.interlaced, self.resampler)




def route_particles(pos_t, mass_t, pm, dmin, dmax):
    """Duplicate each particle to every rank whose x-slab any of its
    deposit cells falls in (the pmesh decompose/exchange step, reference
    :271-284, with ghost width from the window support and interlacing
    shift), then alltoall the payloads."""
    import torch
    comm = pm.comm
    ws = comm.size
    invH0 = float(pm.Nmesh[0]) / float(pm.BoxSize[0])
    fu = torch.floor(pos_t[:, 0] * invH0).long()
    n0 = int(pm.Nmesh[0])
    nx_l = pm.nx_local

    if dmax - dmin < nx_l:
        # The ghost window spans (dmax - dmin) cells < one slab, so a
        # particle's destination set is ONE or TWO (wrap-adjacent)
        # ranks: dedup analytically instead of torch.unique over all
        # (particle, rank) pairs — the sort-based unique plus a second
        # full argsort cost ~2 large device sorts per chunk per step,
        # which would dominate the per-rank step at 8 GPUs.
        r_lo = torch.div(torch.remainder(fu + dmin, n0), nx_l,
                         rounding_mode='floor')
        r_hi = torch.div(torch.remainder(fu + dmax, n0), nx_l,
                         rounding_mode='floor')
        two = r_hi != r_lo
        ar = torch.arange(len(pos_t), device=pos_t.device)
        idxs = torch.cat([ar, ar[two]])
        ranks = torch.cat([r_lo, r_hi[two]])
    else:
        # degenerate slabs (nx_local smaller than the stencil): the
        # generic per-offset route with pair dedup
        idx_list = []
        rank_list = []
        for d in range(dmin, dmax + 1):
            cell = torch.remainder(fu + d, n0)
            rank_list.append(torch.div(cell, nx_l,
                                       rounding_mode='floor'))
            idx_list.append(torch.arange(len(pos_t),
                                         device=pos_t.device))
        keys = torch.unique(torch.cat(idx_list) * ws
                            + torch.cat(rank_list))
        idxs = torch.div(keys, ws, rounding_mode='floor')
        ranks = keys - idxs * ws

    order = torch.argsort(ranks, stable=True)
    idxs = idxs[order]
    ranks = ranks[order]
    counts = torch.bincount(ranks, minlength=ws).cpu().tolist()

    if mass_t is None:
        recv = exchange_particle_arrays(pos_t[idxs], counts, comm)
        return recv.contiguous(), None
    payload = torch.cat([pos_t[idxs], mass_t[idxs, None]], dim=1)
    recv = exchange_particle_arrays(payload, counts, comm)
    return recv[:, :3].contiguous(), recv[:, 3].contiguous()


def paint_raw(pos_t, pm, resampler='cic'):
    """Paint device positions with unit mass into a fresh (unnormalized)
    RealField — the bare ``pm.paint`` the reconstruction driver uses
    (fftrecon.py:159-161).  Handles routing and the locality sort."""
    field = RealField(pm)
    if len(pos_t) == 0:
        return field
    lib = hiplib.require()
    if pm.comm.size > 1:
        dmin, dmax = _GHOST_RANGE[(resampler, False)]
        pos_t, _ = route_particles(pos_t, None, pm, dmin, dmax)
    # the window MUST reach the sort: the pair-bucket pipeline's
    # duplication range is stencil-dependent (a CIC-range sort under a
    # TSC paint would drop group-boundary deposits)
    pos_soa, _, sorted_, rowtab = _prepare_particles(
        pos_t, None, pm, window=resampler, interlaced=False)
    if rowtab is not None:
        if isinstance(rowtab, tuple):
            table, pair_gs, n_eff = rowtab
        else:
            table, pair_gs, n_eff = rowtab, -1, len(pos_t)
        hiplib.check(lib.nbk_paint_gather_f64(
            hiplib.dptr(pos_soa), None, n_eff,
            hiplib.i64_arr(pm.Nmesh), hiplib.f64_arr(pm.BoxSize),
            hiplib.WINDOW_IDS[resampler], 0.0, hiplib.dptr(table),
            hiplib.dptr(field.value), pm.x_start, pm.nx_local, 0,
            pair_gs, hiplib.cur_stream()), 'nbk_paint_gather_f64')
        return field
    paint_fn = lib.nbk_paint_sorted_f64 if sorted_ else lib.nbk_paint_f64
    hiplib.check(paint_fn(
        hiplib.dptr(pos_soa), None, len(pos_t),
        hiplib.i64_arr(pm.Nmesh), hiplib.f64_arr(pm.BoxSize),
        hiplib.WINDOW_IDS[resampler], 0.0, hiplib.dptr(field.value),
        pm.x_start, pm.nx_local, hiplib.cur_stream()), 'nbk_paint')
    return field

    # -- compensation actions (reference :405-451) ------------------------

def get_compensation(interlaced, resampler):
    """(mode, filter, kind) action for the window compensation
    (reference :419-451)."""
    if interlaced:
        d = {'cic': CompensateCIC, 'tsc': CompensateTSC,
             'pcs': CompensatePCS}
    else:
        d = {'cic': CompensateCICShotnoise, 'tsc': CompensateTSCShotnoise,
             'pcs': CompensatePCSShotnoise}
    if resampler not in d:
        raise ValueError("compensation for window %s is not defined"
                         % resampler)
    return [('complex', d[resampler], 'circular')]


# ---- the six compensation filters (reference :453-594) -----------------
# w is the list of circular coordinates in [-pi, pi); these run on host
# coordinate arrays when called directly, and are dispatched by identity
# to the nbk_compensate_f64 kernel inside ComplexField.apply.

def _sinc_power(w, v, p):
    for i in range(3):
        wi = w[i]
        tmp = numpy.sinc(0.5 * wi / numpy.pi) ** p
        v = v / tmp
    return v


def CompensateCIC(w, v):
    """Inverse CIC window (Jing 2005 eq. 18, p=2; reference :499-521)."""
    for i in range(3):
        wi = w[i]
        tmp = numpy.sinc(0.5 * wi / numpy.pi) ** 2
        tmp[wi == 0.] = 1.
        v = v / tmp
    return v


def CompensateTSC(w, v):
    """Inverse TSC window (p=3; reference :453-474)."""
    return _sinc_power(w, v, 3)


def CompensatePCS(w, v):
    """Inverse PCS window (p=4; reference :476-497)."""
    return _sinc_power(w, v, 4)


def CompensateCICShotnoise(w, v):
    """CIC with first-order aliasing correction (Jing eq. 20;
    reference :573-594)."""
    for i in range(3):
        wi = w[i]
        v = v / (1 - 2. / 3 * numpy.sin(0.5 * wi) ** 2) ** 0.5
    return v


def CompensateTSCShotnoise(w, v):
    """TSC with first-order aliasing correction (reference :523-545)."""
    for i in range(3):
        wi = w[i]
        s = numpy.sin(0.5 * wi) ** 2
        v = v / (1 - s + 2. / 15 * s ** 2) ** 0.5
    return v


def CompensatePCSShotnoise(w, v):
    """PCS with first-order aliasing correction (reference :547-571)."""
    for i in range(3):
        wi = w[i]
        s = numpy.sin(0.5 * wi) ** 2
        v = v / (1 - 4. / 3. * s + 2. / 5. * s ** 2
                 - 4. / 315. * s ** 3) ** 0.5
    return v


_COMPENSATION_KERNELS = {
    CompensateCIC: ('cic', True),
    CompensateTSC: ('tsc', True),
    CompensatePCS: ('pcs', True),
    CompensateCICShotnoise: ('cic', False),
    CompensateTSCShotnoise: ('tsc', False),
    CompensatePCSShotnoise: ('pcs', False),
}


def lookup_compensation(func):
    """(window, interlaced) if ``func`` is a built-in compensation filter
    with a HIP kernel, else None."""
    return _COMPENSATION_KERNELS.get(func)
