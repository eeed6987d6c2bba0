"""
ConvolvedFFTPower — survey-geometry power spectrum multipoles of an FKP
density field (reference nbodykit/algorithms/convpower/fkp.py:75-808;
the Hand et al. 2017 estimator: spherical-harmonic addition theorem, so
each multipole costs 2l+1 FFTs — Bianchi et al. 2015 / Scoccimarro 2015
lineage).

Per multipole l > 0:
    A_l(k) = 4 pi / (2l+1) sum_m Y_lm(khat) FFT[ Y_lm(xhat) F(x) ]
    P_l(k) = (2l+1) / A_norm * < A_0(k) A_l(k)* >_{k-shell}
with F(x) the FKP density painted by FKPCatalogMesh and A_norm the
randoms-based normalization (Beutler et al. 2014 eqs. 13-15).
All meshes and FFTs run on the GPU; the Y_lm factors are sympy-generated
polynomials in the unit coordinates evaluated directly on torch grids.
"""
import logging
import time

import numpy

from nbodykit_amd import CurrentMPIComm
from nbodykit_amd.utils import timer
from nbodykit_amd.binned_statistic import BinnedStatistic
from nbodykit_amd.algorithms.fftpower import (project_to_basis,
                                              _find_unique_edges)
from nbodykit_amd.pm import ComplexField, RealField


def get_real_Ylm(l, m):
    """Real spherical harmonic Y_lm as a polynomial in the unit-vector
    components (reference :12-73; sympy-derived, numpy-lambdified —
    numeric constants fold, so the function also accepts torch tensors)."""
    import sympy as sp

    l = int(l)
    m = int(m)

    x, y, z, r = sp.symbols('x y z r', real=True, positive=True)
    xhat, yhat, zhat = sp.symbols('xhat yhat zhat', real=True,
                                  positive=True)
    phi, theta = sp.symbols('phi theta')
    defs = [(sp.sin(phi), y / sp.sqrt(x ** 2 + y ** 2)),
            (sp.cos(phi), x / sp.sqrt(x ** 2 + y ** 2)),
            (sp.cos(theta), z / sp.sqrt(x ** 2 + y ** 2 + z ** 2))]

    if m == 0:
        amp = sp.sqrt((2 * l + 1) / (4 * numpy.pi))
    else:
        amp = sp.sqrt(2 * (2 * l + 1) / (4 * numpy.pi)
                      * sp.factorial(l - abs(m))
                      / sp.factorial(l + abs(m)))

    expr = (-1) ** m * sp.assoc_legendre(l, abs(m), sp.cos(theta))
    if m < 0:
        expr *= sp.expand_trig(sp.sin(abs(m) * phi))
    elif m > 0:
        expr *= sp.expand_trig(sp.cos(m * phi))

    expr = sp.together(expr.subs(defs)).subs(x ** 2 + y ** 2 + z ** 2,
                                             r ** 2)
    expr = amp * expr.expand().subs([(x / r, xhat), (y / r, yhat),
                                     (z / r, zhat)])
    Ylm = sp.lambdify((xhat, yhat, zhat), expr, 'numpy')

    Ylm.expr = expr
    Ylm.l = l
    Ylm.m = m
    return Ylm


class ConvolvedFFTPower(object):
    logger = logging.getLogger('ConvolvedFFTPower')

    def __init__(self, first, poles, second=None, Nmesh=None, kmin=0.,
                 kmax=None, dk=None, use_fkp_weights=None, P0_FKP=None):
        if use_fkp_weights is not None or P0_FKP is not None:
            raise ValueError(
                "use_fkp_weights and P0_FKP are deprecated. Assign a "
                "FKPWeight column with FKPWeightFromNbar(nbar)")

        first = _cast_mesh(first, Nmesh=Nmesh)
        if second is not None:
            second = _cast_mesh(second, Nmesh=Nmesh)
        else:
            second = first

        if not is_valid_crosscorr(first, second):
            raise NotImplementedError(
                "ConvolvedFFTPower cross-correlations currently require "
                "the same FKPCatalog (data/randoms), such that only the "
                "weight column can vary")

        self.first = first
        self.second = second
        self.comm = first.comm
        assert second.comm is first.comm

        if numpy.isscalar(poles):
            poles = [poles]

        self.attrs = {}
        self.attrs['poles'] = poles
        self.attrs['dk'] = dk
        self.attrs['kmin'] = kmin
        self.attrs['kmax'] = kmax
        self.attrs['Nmesh'] = self.first.attrs['Nmesh'].copy()
        self.attrs['BoxSize'] = self.first.attrs['BoxSize']
        self.attrs['BoxPad'] = self.first.attrs['BoxPad']
        self.attrs['BoxCenter'] = self.first.attrs['BoxCenter']
        self.attrs['mesh.resampler'] = self.first.resampler
        self.attrs['mesh.interlaced'] = self.first.interlaced

        self.run()

    def run(self):
        pm = self.first.pm

        dk = 2 * numpy.pi / pm.BoxSize.min() if self.attrs['dk'] is None \
            else self.attrs['dk']
        kmin = self.attrs['kmin']
        kmax = self.attrs['kmax']
        if kmax is None:
            kmax = numpy.pi * pm.Nmesh.min() / pm.BoxSize.max() + dk / 2

        if dk > 0:
            kedges = numpy.arange(kmin, kmax, dk)
            kcoords = None
        else:
            from nbodykit_amd.pm import _int_freqs
            N = [int(n) for n in pm.Nmesh]
            k0 = 2 * numpy.pi / pm.BoxSize
            fz = numpy.arange(N[2] // 2 + 1, dtype='f8')
            if N[2] % 2 == 0:
                fz[-1] = -(N[2] // 2)
            k = [(_int_freqs(N[0]) * k0[0]).reshape(-1, 1, 1),
                 (_int_freqs(N[1]) * k0[1]).reshape(1, -1, 1),
                 (fz * k0[2]).reshape(1, 1, -1)]
            kedges, kcoords = _find_unique_edges(
                k, 2 * numpy.pi / pm.BoxSize, kmax, pm.comm)

        result = self._compute_multipoles(kedges)

        self.poles = BinnedStatistic(['k'], [kedges], result,
                                     fields_to_sum=['modes'],
                                     coords=[kcoords], **self.attrs)
        self.edges = kedges

    # -- core -------------------------------------------------------------
    def _compute_multipoles(self, kedges):
        import torch
        comm = self.comm
        rank = comm.rank
        pm = self.first.pm

        for source in [self.first, self.second]:
            source.actions[:] = []
            source.compensated = False

        compensation = {}
        for name, mesh in zip(['first', 'second'],
                              [self.first, self.second]):
            compensation[name] = get_compensation(mesh)

        muedges = numpy.linspace(-1, 1, 2, endpoint=True)
        edges = [kedges, muedges]

        poles = sorted(self.attrs['poles'])
        cols = ['k'] + ['power_%d' % l for l in poles] + ['modes']
        dtype = numpy.dtype([(c, 'f8' if c == 'k' else
                              ('i8' if c == 'modes' else 'c8'))
                             for c in cols])
        result = numpy.empty(len(kedges) - 1, dtype=dtype)

        # original-coordinate offset for xhat (cell centres;
        # reference :455-458)
        offset = self.attrs['BoxCenter'] + 0.5 * pm.BoxSize / pm.Nmesh

        if 0 not in poles:
            poles = [0] + poles
        assert poles[0] == 0

        Ylms = [[get_real_Ylm(l, m) for m in range(-l, l + 1)]
                for l in poles[1:]]

        # paint the FKP density of the first mesh
        rfield1 = self.first.compute(Nmesh=self.attrs['Nmesh'])
        meta1 = rfield1.attrs.copy()
        self.attrs['alpha'] = meta1['alpha']

        cfield = rfield1.r2c()
        if compensation['first'] is not None:
            cfield.apply(out=Ellipsis, **compensation['first'])
        if rank == 0:
            self.logger.info('ell = 0 done; 1 r2c completed')

        volume = float(pm.BoxSize.prod())
        A0_1 = cfield.value * volume

        if self.first is not self.second:
            rfield2 = self.second.compute(Nmesh=self.attrs['Nmesh'])
            meta2 = rfield2.attrs.copy()
            if 0 in self.attrs['poles']:
                A0_2f = rfield2.r2c()
                if compensation['second'] is not None:
                    A0_2f.apply(out=Ellipsis, **compensation['second'])
                A0_2 = A0_2f.value * volume
        else:
            rfield2 = rfield1
            meta2 = meta1
            if 0 in self.attrs['poles']:
                A0_2 = A0_1

        if not numpy.allclose(meta1['alpha'], meta2['alpha'], rtol=1e-3):
            raise ValueError("different ``alpha`` values found for "
                             "first/second meshes")

        density2 = rfield2.value.clone()

        # normalization (Scoccimarro 2015 eq. 49; reference :540-570)
        for name in ['data', 'randoms']:
            self.attrs[name + '.norm'] = self.normalization(
                name, self.attrs['alpha'])

        if self.attrs['randoms.norm'] > 0:
            norm = 1.0 / self.attrs['randoms.norm']
            Adata = self.attrs['data.norm']
            Aran = self.attrs['randoms.norm']
            if not numpy.allclose(Adata, Aran, rtol=0.05):
                raise ValueError(
                    "normalization in ConvolvedFFTPower different by more "
                    "than 5%%: randoms.norm = %.6f, data.norm = %.6f; the "
                    "n(z) columns should be normalized to the data n(z)"
                    % (Aran, Adata))
        else:
            norm = 1.0
            if rank == 0:
                self.logger.info("normalization neglected (no randoms)")

        # unit-coordinate grids (torch, on device)
        def unit_grids(coords, off):
            comps = [torch.as_tensor(c).to('cuda') + o
                     for c, o in zip(coords, off)]
            norm2 = sum((c ** 2 for c in comps))
            norm_ = torch.sqrt(norm2)
            norm_ = torch.where(norm_ == 0, torch.inf, norm_)
            return [c / norm_ for c in comps]

        xhat = unit_grids(rfield2.x, offset)
        khat = unit_grids(cfield.x, (0., 0., 0.))

        start = time.time()
        for iell, ell in enumerate(poles[1:]):
            Aell = torch.zeros_like(cfield.value)

            for Ylm in Ylms[iell]:
                yx = Ylm(xhat[0], xhat[1], xhat[2])
                rf2 = RealField(pm, tensor=density2 * yx)
                cf = rf2.r2c()
                yk = Ylm(khat[0], khat[1], khat[2])
                Aell += cf.value * yk
                if rank == 0:
                    self.logger.debug("done term for Y(l=%d, m=%d)"
                                      % (Ylm.l, Ylm.m))

            Aell_f = ComplexField(pm, tensor=Aell)
            if compensation['second'] is not None:
                Aell_f.apply(out=Ellipsis, **compensation['second'])
            # 4 pi from the addition theorem + the volume factor
            Aell_f.value.mul_(4 * numpy.pi * volume)
            if rank == 0:
                self.logger.info('ell = %d done; %s r2c completed'
                                 % (ell, len(Ylms[iell])))

            Aell_f.value.copy_(norm * A0_1 * Aell_f.value.conj())
            proj_result, _ = project_to_basis(Aell_f, edges)
            result['power_%d' % ell][:] = numpy.squeeze(proj_result[2])

        if rank == 0 and len(poles) > 1:
            self.logger.info("higher order multipoles computed in "
                             "elapsed time %s" % timer(start, time.time()))

        if 0 in self.attrs['poles']:
            P0 = ComplexField(pm, tensor=norm * A0_1 * A0_2.conj())
            proj_result, _ = project_to_basis(P0, edges)
            result['power_0'][:] = numpy.squeeze(proj_result[2])

        result['k'][:] = numpy.squeeze(proj_result[0])
        result['modes'][:] = numpy.squeeze(proj_result[-1])

        self.attrs['shotnoise'] = self.shotnoise(self.attrs['alpha'])

        if self.first is self.second:
            copy_meta(self.attrs, meta1)
        else:
            copy_meta(self.attrs, meta1, prefix='first')
            copy_meta(self.attrs, meta2, prefix='second')
        return result

    # -- normalization / shot noise (reference :657-760) ------------------
    def normalization(self, name, alpha):
        assert name in ['data', 'randoms']
        if name + '.norm' not in self.attrs:
            src = self.first.source[name]
            sel = numpy.asarray(src[self.first.selection], dtype=bool)
            comp_weight = numpy.asarray(
                src[self.first.comp_weight])[sel]
            nbar = numpy.asarray(
                self.second.source[name][self.second.nbar])[sel]
            fkp1 = numpy.asarray(src[self.first.fkp_weight])[sel]
            if self.second is self.first:
                fkp2 = fkp1
            else:
                fkp2 = numpy.asarray(
                    self.second.source[name][self.second.fkp_weight])[sel]
            A = float((nbar * comp_weight * fkp1 * fkp2).sum())
            if name == 'randoms':
                A *= alpha
            self.attrs[name + '.norm'] = self.comm.allreduce(A)
        return self.attrs[name + '.norm']

    def shotnoise(self, alpha):
        if 'shotnoise' in self.attrs:
            return self.attrs['shotnoise']
        Pshot = 0
        for name in ['data', 'randoms']:
            src = self.first.source[name]
            sel = numpy.asarray(src[self.first.selection], dtype=bool)
            comp_weight = numpy.asarray(
                src[self.first.comp_weight])[sel]
            fkp1 = numpy.asarray(src[self.first.fkp_weight])[sel]
            if self.first is self.second:
                fkp2 = fkp1
            else:
                fkp2 = numpy.asarray(
                    self.second.source[name][self.second.fkp_weight])[sel]
            S = float((comp_weight ** 2 * fkp1 * fkp2).sum())
            if name == 'randoms':
                S *= alpha ** 2
            Pshot += S
        Pshot = self.comm.allreduce(Pshot)
        return Pshot / self.attrs['randoms.norm']

    # -- conversion + io (reference :282-406) ------------------------------
    def to_pkmu(self, mu_edges, max_ell):
        """Invert the measured multipoles into P(k, mu) wedges
        (reference :282-338)."""
        from scipy.special import legendre

        def compute_coefficient(ell, mumin, mumax):
            norm = 1.0 / (mumax - mumin)
            c = legendre(ell).integ()
            return norm * (c(mumax) - c(mumin))

        ells = sorted([ell for ell in self.attrs['poles']
                       if ell <= max_ell])
        wedges = []
        for imu in range(len(mu_edges) - 1):
            mumin, mumax = mu_edges[imu], mu_edges[imu + 1]
            pkmu = sum(compute_coefficient(ell, mumin, mumax)
                       * self.poles['power_%d' % ell] for ell in ells)
            wedges.append(pkmu)
        data = numpy.empty((len(self.poles['k']), len(wedges)),
                           dtype=[('power', 'c8'), ('k', 'f8'),
                                  ('mu', 'f8')])
        for imu, w in enumerate(wedges):
            data['power'][:, imu] = w
            data['k'][:, imu] = self.poles['k']
            data['mu'][:, imu] = 0.5 * (mu_edges[imu] + mu_edges[imu + 1])
        return BinnedStatistic(['k', 'mu'],
                               [self.poles.edges['k'], mu_edges], data)

    def __getstate__(self):
        return dict(edges=self.edges,
                    poles=self.poles.data,
                    attrs=self.attrs)

    def __setstate__(self, state):
        self.__dict__.update(state)
        self.poles = BinnedStatistic(['k'], [self.edges], self.poles,
                                     fields_to_sum=['modes'])

    def save(self, output):
        import json
        from nbodykit_amd.utils import JSONEncoder
        state = self.__getstate__()
        if self.comm.rank == 0:
            with open(output, 'w') as ff:
                json.dump(state, ff, cls=JSONEncoder)

    @classmethod
    @CurrentMPIComm.enable
    def load(cls, output, comm=None):
        import json
        from nbodykit_amd.utils import JSONDecoder
        if comm.rank == 0:
            with open(output, 'r') as ff:
                state = json.load(ff, cls=JSONDecoder)
        else:
            state = None
        state = comm.bcast(state)
        self = object.__new__(cls)
        self.__setstate__(state)
        self.comm = comm
        return self


def _cast_mesh(mesh, Nmesh):
    from .catalog import FKPCatalog
    from .catalogmesh import FKPCatalogMesh
    if not isinstance(mesh, (FKPCatalogMesh, FKPCatalog)):
        raise TypeError("input sources should be a FKPCatalog or "
                        "FKPCatalogMesh")
    if isinstance(mesh, FKPCatalog):
        mesh = mesh.to_mesh(Nmesh=Nmesh, dtype='f8', compensated=False)
    if Nmesh is not None and any(mesh.attrs['Nmesh'] != Nmesh):
        raise ValueError("Mismatched Nmesh between __init__ and "
                         "mesh.attrs; specify `Nmesh` to to_mesh()")
    return mesh


def get_compensation(mesh):
    try:
        compensation = mesh._get_compensation()
        return {'func': compensation[0][1], 'kind': compensation[0][2]}
    except ValueError:
        return None


def copy_meta(attrs, meta, prefix=""):
    if prefix:
        prefix += '.'
    for key in meta:
        if key.startswith('data.') or key.startswith('randoms.'):
            attrs[prefix + key] = meta[key]


def is_valid_crosscorr(first, second):
    if second.source is not first.source:
        return False
    same_cols = ['selection', 'comp_weight', 'nbar']
    if any(getattr(second, name) != getattr(first, name)
           for name in same_cols):
        return False
    return True
