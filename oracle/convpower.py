"""
ConvolvedFFTPower restatement (reference
nbodykit/algorithms/convpower/fkp.py:408-760 and catalogmesh.py:124-204)
— numpy, single process, with the real spherical harmonics built from
scipy.special.sph_harm (an INDEPENDENT formulation from the product's
sympy-generated polynomials, so the two cross-validate).
"""
import numpy

from .mesh import MeshGeometry, r2c, real_coords, complex_coords
from .paint import paint
from .fftpower import apply_compensation, project_to_basis


def _sph(m, l, azimuth, polar):
    """Complex Y_lm; scipy >= 1.15 renames sph_harm -> sph_harm_y with
    swapped argument order."""
    try:
        from scipy.special import sph_harm_y
        return sph_harm_y(l, m, polar, azimuth)
    except ImportError:
        from scipy.special import sph_harm
        return sph_harm(m, l, azimuth, polar)


def real_Ylm(l, m, xh, yh, zh):
    """Real spherical harmonic from the complex scipy ones
    (https://en.wikipedia.org/wiki/Spherical_harmonics#Real_form)."""
    theta = numpy.arctan2(yh, xh)            # azimuth
    phi = numpy.arccos(numpy.clip(zh, -1, 1))  # polar
    if m > 0:
        return numpy.sqrt(2) * (-1) ** m * _sph(m, l, theta, phi).real
    if m < 0:
        return numpy.sqrt(2) * (-1) ** m * _sph(-m, l, theta, phi).imag
    return _sph(0, l, theta, phi).real


def fkp_density(data_pos, ran_pos, geom, box_center, alpha,
                data_w=None, ran_w=None, resampler='cic'):
    """F(x) = [W_data - alpha W_randoms] / V_cell on the recentered box
    (reference catalogmesh.py:124-204).  Weights are comp*fkp totals.
    Positions are shifted by BoxCenter only (range [-L/2, L/2]; the
    paint wraps negative cells) so cell indexing matches the product's
    fftfreq-style 'relative' coordinate convention."""
    mesh = numpy.zeros(tuple(int(x) for x in geom.Nmesh))
    paint(data_pos - box_center, 1.0 if data_w is None else data_w, mesh,
          geom, resampler=resampler)
    mesh_r = numpy.zeros_like(mesh)
    paint(ran_pos - box_center, 1.0 if ran_w is None else ran_w, mesh_r,
          geom, resampler=resampler)
    mesh -= alpha * mesh_r
    mesh /= float(numpy.prod(geom.H))
    return mesh


def convpower_oracle(data_pos, ran_pos, poles, Nmesh, BoxSize, BoxCenter,
                     nbar_data, nbar_ran, data_comp=None, ran_comp=None,
                     data_fkp=None, ran_fkp=None, resampler='cic',
                     compensated=False, dk=None, kmin=0., kmax=None):
    """
    Survey multipoles of the FKP field.  ``nbar_*`` are per-object n(z)
    arrays; comp/fkp weights default to 1.  Returns a dict with k,
    power_ell (complex64 like the reference), modes, attrs.

    """
    geom = MeshGeometry(Nmesh, BoxSize, dtype='f8')
    BoxCenter = numpy.asarray(BoxCenter, dtype='f8')

    n_d = len(data_pos)
    n_r = len(ran_pos)
    ones = lambda n: numpy.ones(n)
    data_comp = ones(n_d) if data_comp is None else numpy.asarray(data_comp)
    ran_comp = ones(n_r) if ran_comp is None else numpy.asarray(ran_comp)
    data_fkp = ones(n_d) if data_fkp is None else numpy.asarray(data_fkp)
    ran_fkp = ones(n_r) if ran_fkp is None else numpy.asarray(ran_fkp)

    W_data = data_comp.sum()
    W_ran = ran_comp.sum()
    alpha = W_data / W_ran

    F = fkp_density(data_pos, ran_pos, geom, BoxCenter, alpha,
                    data_w=data_comp * data_fkp,
                    ran_w=ran_comp * ran_fkp, resampler=resampler)

    volume = float(numpy.prod(geom.BoxSize))
    A0 = r2c(F, geom)
    if compensated:
        apply_compensation(A0, geom, resampler, False)
    A0 = A0 * volume

    # normalization + shot noise (fkp.py:657-760)
    norm_ran = alpha * (nbar_ran * ran_comp * ran_fkp ** 2).sum()
    norm_data = (nbar_data * data_comp * data_fkp ** 2).sum()
    norm = 1.0 / norm_ran if norm_ran > 0 else 1.0
    Pshot = ((data_comp ** 2 * data_fkp ** 2).sum()
             + alpha ** 2 * (ran_comp ** 2 * ran_fkp ** 2).sum())
    shotnoise = Pshot / norm_ran if norm_ran > 0 else numpy.nan

    if dk is None:
        dk = 2 * numpy.pi / geom.BoxSize.min()
    if kmax is None:
        kmax = numpy.pi * geom.Nmesh.min() / geom.BoxSize.max() + dk / 2
    kedges = numpy.arange(kmin, kmax, dk)
    muedges = numpy.linspace(-1, 1, 2, endpoint=True)

    # unit coordinate grids; offsets restore the original sky frame at
    # cell centres (fkp.py:455-458, 528-537)
    offset = BoxCenter + 0.5 * geom.BoxSize / geom.Nmesh
    xg = [c + o for c, o in zip(real_coords(geom), offset)]
    xnorm = numpy.sqrt(sum(c ** 2 for c in xg))
    xh = [numpy.broadcast_to(c / numpy.where(xnorm == 0, numpy.inf, xnorm),
                             xnorm.shape) for c in xg]
    kg = complex_coords(geom)
    knorm = numpy.sqrt(sum(c ** 2 for c in kg))
    knorm = numpy.where(knorm == 0, numpy.inf, knorm)
    kh = [numpy.broadcast_to(c / knorm, knorm.shape) for c in kg]

    ells = sorted(poles)
    out = {'attrs': {'alpha': alpha, 'shotnoise': shotnoise,
                     'data.norm': norm_data, 'randoms.norm': norm_ran,
                     'data.W': W_data, 'randoms.W': W_ran},
           'kedges': kedges}

    for ell in ells:
        if ell == 0:
            P = norm * A0 * numpy.conj(A0)
        else:
            Aell = numpy.zeros_like(A0)
            for m in range(-ell, ell + 1):
                yx = real_Ylm(ell, m, xh[0], xh[1], xh[2])
                cf = r2c(F * yx, geom)
                yk = real_Ylm(ell, m, kh[0], kh[1], kh[2])
                Aell += cf * yk
            if compensated:
                apply_compensation(Aell, geom, resampler, False)
            Aell *= 4 * numpy.pi * volume
            P = norm * A0 * numpy.conj(Aell)

        result, _ = project_to_basis(P, geom, [kedges, muedges])
        xmean, mumean, y2d, N2d = result
        out['power_%d' % ell] = numpy.squeeze(y2d).astype('c8')
        out['k'] = numpy.squeeze(xmean)
        out['modes'] = numpy.squeeze(N2d)
    return out
