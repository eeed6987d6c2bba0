"""
FFTRecon — FFT-based Lagrangian reconstruction in a periodic box
(reference nbodykit/algorithms/fftrecon.py:11-269; Eisenstein et al 2007
/ Schmittfull et al 2015 schemes LGS, LF2, LRR).

Flow (reference :132-269): paint the unshifted data overdensity
delta_d = counts/nbar, solve the smoothed Zel'dovich displacement per
axis in k-space (``nbk_recon_displacement_f64``:
i k_d/k^2 exp(-k^2 R^2/2) / (bias (1 + f/bias mu^2))), read it out at
the particle positions (CIC ``RealField.readout``), shift data (and
optionally randoms), and combine the shifted paints per scheme.  The
result is a MeshSource whose real field is delta (counts/nbar - paired
differences; NOT 1+delta — reference :215 FIXME), so
``FFTPower(FFTRecon(...), ...)`` works directly.
"""
import logging
import warnings

import numpy

from nbodykit_amd import hiplib
from nbodykit_amd.base.catalog import CatalogSource
from nbodykit_amd.base.mesh import MeshSource
from nbodykit_amd.source.mesh.catalog import paint_raw


class FFTRecon(MeshSource):
    logger = logging.getLogger('FFTRecon')

    def __init__(self, data, ran, Nmesh, bias=1.0, f=0.0, los=[0, 0, 1],
                 R=20, position='Position', revert_rsd_random=False,
                 scheme='LGS', BoxSize=None):
        assert scheme in ['LGS', 'LF2', 'LRR']
        assert isinstance(data, CatalogSource)
        assert isinstance(ran, CatalogSource)

        comm = data.comm
        assert data.comm == ran.comm

        if Nmesh is None:
            Nmesh = data.attrs['Nmesh']
        if BoxSize is None:
            BoxSize = data.attrs['BoxSize']

        los = numpy.array(los, dtype='f8', copy=True)
        los /= (los ** 2).sum()
        assert len(los) == 3
        assert (~numpy.isnan(los)).all()

        MeshSource.__init__(self, comm, Nmesh, BoxSize, 'f8')

        if (self.pm.BoxSize / self.pm.Nmesh).max() > R:
            if comm.rank == 0:
                warnings.warn("The smoothing radius smaller than the mesh "
                              "cell size. This may produce undesired "
                              "numerical results.")

        assert position in data.columns
        assert position in ran.columns
        self.position = position

        self.attrs['bias'] = bias
        self.attrs['f'] = f
        self.attrs['los'] = los
        self.attrs['R'] = R
        self.attrs['scheme'] = scheme
        self.attrs['revert_rsd_random'] = bool(revert_rsd_random)

        self.data = data
        self.ran = ran

        if self.comm.rank == 0:
            self.logger.info(
                "Reconstruction for bias=%g, f=%g, smoothing R=%g los=%s"
                % (bias, f, R, str(los)))
            self.logger.info("Reconstruction scheme = %s" % scheme)

    # -- MeshSource interface ---------------------------------------------
    def to_real_field(self, out=None, normalize=True):
        return self.run()

    def run(self):
        s_d, s_r = self._compute_s()
        return self._helper_paint(s_d, s_r)

    # -- internals ----------------------------------------------------------
    def _positions(self, cat):
        import torch
        col = cat[self.position]
        if isinstance(col, torch.Tensor):
            return col.to(device='cuda', dtype=torch.float64).contiguous()
        arr = numpy.ascontiguousarray(numpy.asarray(col), dtype='f8')
        return torch.as_tensor(arr).to('cuda')

    def work_with(self, pos_t, s, csize):
        """delta = paint(pos - s) / nbar (reference :140-164); shifted
        positions may leave the box — the paint kernel wraps cells."""
        dpos = pos_t if s is None else (pos_t - s)
        delta = paint_raw(dpos, self.pm, resampler='cic')
        nbar = 1.0 * csize / float(numpy.prod(self.pm.Nmesh))
        delta.value.div_(nbar)
        return delta

    def _summary_field(self, field, name):
        cmean = field.cmean()
        if self.comm.rank == 0:
            self.logger.info("painted %s, mean=%g" % (name, cmean))

    def _helper_paint(self, s_d, s_r):
        pos_d = self._positions(self.data)
        pos_r = self._positions(self.ran)

        delta_s_r = self.work_with(pos_r, s_r, self.ran.csize)
        self._summary_field(delta_s_r, "delta_s_r (shifted)")

        def LGS():
            delta_s_d = self.work_with(pos_d, s_d, self.data.csize)
            self._summary_field(delta_s_d, "delta_s_d (shifted)")
            delta_s_d.value.sub_(delta_s_r.value)
            return delta_s_d

        def LRR():
            delta_s_nr = self.work_with(pos_r, -s_r, self.ran.csize)
            self._summary_field(delta_s_nr, "delta_s_nr (reverse shifted)")
            delta_d = self.work_with(pos_d, None, self.data.csize)
            self._summary_field(delta_d, "delta_d (unshifted)")
            delta_s_nr.value.add_(delta_s_r.value).mul_(0.5)
            delta_d.value.sub_(delta_s_nr.value)
            return delta_d

        scheme = self.attrs['scheme']
        if scheme == 'LGS':
            delta_recon = LGS()
        elif scheme == 'LRR':
            delta_recon = LRR()
        else:  # LF2 (reference :195-201)
            lgs = LGS()
            lrr = LRR()
            lgs.value.mul_(3.0 / 7.0)
            lrr.value.mul_(4.0 / 7.0)
            lgs.value.add_(lrr.value)
            delta_recon = lgs

        self._summary_field(delta_recon, "delta_recon")
        return delta_recon

    def _compute_s(self):
        """Solve the reconstruction displacements (reference :219-269)."""
        import torch
        lib = hiplib.require()
        pm = self.pm

        pos_d = self._positions(self.data)
        pos_r = self._positions(self.ran)

        delta_d = self.work_with(pos_d, None, self.data.csize)
        self._summary_field(delta_d, "delta_d (unshifted)")
        delta_k = delta_d.r2c(out=Ellipsis)

        nmesh = hiplib.i64_arr(pm.Nmesh)
        box = hiplib.f64_arr(pm.BoxSize)
        los = self.attrs['los']
        stream = hiplib.cur_stream()

        def solve_displacement(pos_t):
            s = torch.zeros((len(pos_t), 3), dtype=torch.float64,
                            device='cuda')
            disp_k = delta_k.copy()
            for d in range(3):
                hiplib.check(lib.nbk_recon_displacement_f64(
                    hiplib.dptr(disp_k.value), hiplib.dptr(delta_k.value),
                    nmesh, box, hiplib.i64_arr(delta_k.dims),
                    hiplib.i64_arr(delta_k.off), d,
                    float(self.attrs['R']), float(self.attrs['bias']),
                    float(self.attrs['f']), hiplib.f64_arr(los), stream),
                    'nbk_recon_displacement_f64')
                disp = disp_k.c2r()
                s[:, d] = disp.readout(pos_t, resampler='cic')
            return s

        s_d = solve_displacement(pos_d)
        s_d_std = numpy.asarray(self.comm.allreduce(
            (s_d ** 2).sum(dim=0).cpu().numpy())) / self.data.csize
        if self.comm.rank == 0:
            self.logger.info("Solved displacements of data, std(s_d) = %s"
                             % str(s_d_std ** 0.5))

        s_r = solve_displacement(pos_r)

        los_t = torch.as_tensor(los).to('cuda')
        f = self.attrs['f']
        # shifting conventions (reference :260-267)
        s_d *= (1 + los_t * f)
        if self.attrs['revert_rsd_random']:
            s_r *= (1 + los_t * f)
        return s_d, s_r
