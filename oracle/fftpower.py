"""
FFTPower restatement: window compensation (nbodykit/source/mesh/
catalog.py:419-594), 3D power (algorithms/fftpower.py:91-143), basis
projection (fftpower.py:507-701 with the SlabIterator/Hermitian-weight
semantics of meshtools.py), k-edge construction (fftpower.py:284-299,
732-769), and a single-process end-to-end driver used as the parity
oracle and the CPU baseline.
"""
import numpy
from scipy.special import legendre

from .mesh import (MeshGeometry, r2c, c2r, complex_coords,
                   complex_circular_coords, real_coords)
from .catalogmesh import to_real_field


# ---------------------------------------------------------------- filters

def compensation_filter(w, resampler, interlaced):
    """
    The Fourier-space division factor for one mode grid, as a full
    (Nx,Ny,Nzh) array.  ``w`` is the circular-coordinate list in [-pi,pi).

    interlaced=True -> plain inverse-window (Jing 2005 eq. 18: sinc^-p;
    source/mesh/catalog.py:453-521); interlaced=False -> first-order
    aliasing-corrected forms (eq. 20; :523-594).
    """
    out = 1.0
    for wi in w:
        if interlaced:
            p = {'cic': 2, 'tsc': 3, 'pcs': 4}[resampler]
            s = numpy.sinc(0.5 * wi / numpy.pi) ** p
            if resampler == 'cic':
                s = numpy.where(wi == 0., 1., s)  # catalog.py:519
            out = out / s
        else:
            s2 = numpy.sin(0.5 * wi) ** 2
            if resampler == 'cic':
                d = (1 - 2. / 3 * s2) ** 0.5
            elif resampler == 'tsc':
                d = (1 - s2 + 2. / 15 * s2 ** 2) ** 0.5
            elif resampler == 'pcs':
                d = (1 - 4. / 3. * s2 + 2. / 5. * s2 ** 2
                     - 4. / 315. * s2 ** 3) ** 0.5
            else:
                raise ValueError(resampler)
            out = out / d
    return out


def apply_compensation(cfield, geom, resampler, interlaced):
    """In-place compensation of a complex field (the auto-prepended
    'complex'/'circular' action, source/mesh/catalog.py:405-451)."""
    w = complex_circular_coords(geom)
    cfield *= compensation_filter(w, resampler, interlaced)
    return cfield


# ---------------------------------------------------------------- power

def compute_3d_power(c1, c2, geom):
    """p3d = c1 * conj(c2), zero-mode cleared but still binned, times V
    (fftpower.py:114-128)."""
    p3d = c1 * numpy.conj(c2)
    p3d[0, 0, 0] = 0.0
    p3d *= float(numpy.prod(geom.BoxSize))
    return p3d


# ---------------------------------------------------------------- binning

def kedges_linear(geom, dk=None, kmin=0., kmax=None):
    """Default linear k-bin edges (fftpower.py:218-219, 284-293)."""
    if dk is None:
        dk = 2 * numpy.pi / geom.BoxSize.min()
    if kmax is None:
        kmax = numpy.pi * geom.Nmesh.min() / geom.BoxSize.max() + dk / 2
    return numpy.arange(kmin, kmax, dk)


def kedges_unique(geom, kmax=None):
    """dk=0 unique-modulus edges (fftpower.py:732-769, single rank)."""
    x = complex_coords(geom)
    x0 = 2 * numpy.pi / geom.BoxSize
    if kmax is None:
        dk = 2 * numpy.pi / geom.BoxSize.min()
        kmax = numpy.pi * geom.Nmesh.min() / geom.BoxSize.max() + dk / 2

    fx2 = sum(xi ** 2 for xi in x).ravel()

    def unique_binned(values, binning):
        ints = numpy.int64(values / binning + 0.5)
        _, ind = numpy.unique(ints, return_index=True)
        return values[ind]

    fx = unique_binned(fx2, (x0.min() * 0.05) ** 2) ** 0.5
    fx = fx[fx < kmax]
    fx = unique_binned(fx, x0.min() * 1e-5)

    width = numpy.diff(fx)
    edges = fx.copy()
    edges[1:] -= width * 0.5
    edges = numpy.append(edges, [fx[-1] + width[-1] * 0.5])
    edges[0] = 0
    return edges, fx


def redges_unique(geom, rmax):
    """dr=0 unique-separation edges for FFTCorr: the configuration-space
    analogue of ``kedges_unique`` (reference fftcorr.py:96-99 calling
    fftpower.py:732-769 with x = RealField coords and x0 = H)."""
    x = real_coords(geom)
    x0 = geom.BoxSize / geom.Nmesh

    fx2 = sum(xi ** 2 for xi in x).ravel()

    def unique_binned(values, binning):
        ints = numpy.int64(values / binning + 0.5)
        _, ind = numpy.unique(ints, return_index=True)
        return values[ind]

    fx = unique_binned(fx2, (x0.min() * 0.05) ** 2) ** 0.5
    fx = fx[fx < rmax]
    fx = unique_binned(fx, x0.min() * 1e-5)

    width = numpy.diff(fx)
    edges = fx.copy()
    edges[1:] -= width * 0.5
    edges = numpy.append(edges, [fx[-1] + width[-1] * 0.5])
    edges[0] = 0
    return edges, fx


def project_to_basis(y3d, geom, edges, los=(0, 0, 1), poles=(),
                     coords=None, hermitian=True, _return_sums=False):
    """
    Restates fftpower.py:507-701 for a single-process field: iterate
    y-z slabs along axis 0, digitize the coordinate norm^2 and mu, apply
    Hermitian double-count weights along the compressed (last) axis
    (Nyquist/DC excluded: meshtools.py:144-215), Legendre-weight the
    requested multipoles with the odd/even conjugate-pair identity
    (:649-656), bincount into (Nx+2, Nmu+2) arrays, fold the internal
    mu==1 bin into the last visible bin (:674-679).

    Default coords are the compressed wavenumber grid (FFTPower);
    ``coords=real_coords(geom), hermitian=False`` bins a real
    configuration-space field (FFTCorr, fftcorr.py:150-176).

    Returns (result, pole_result) with the same contents as the reference.
    """
    xedges, muedges = edges
    x2edges = xedges ** 2
    Nx = len(xedges) - 1
    Nmu = len(muedges) - 1

    poles = list(poles)
    do_poles = len(poles) > 0
    _poles = [0] + sorted(poles) if 0 not in poles else sorted(poles)
    legpoly = [legendre(ell) for ell in _poles]
    ell_idx = [_poles.index(ell) for ell in poles]
    Nell = len(_poles)
    if any(ell < 0 for ell in _poles):
        raise ValueError("multipole numbers must be non-negative integers")

    musum = numpy.zeros((Nx + 2, Nmu + 2))
    xsum = numpy.zeros((Nx + 2, Nmu + 2))
    ysum = numpy.zeros((Nell, Nx + 2, Nmu + 2), dtype=y3d.dtype)
    Nsum = numpy.zeros((Nx + 2, Nmu + 2), dtype='i8')

    if coords is None:
        coords = complex_coords(geom)
    cy = numpy.take(coords[1], 0, axis=0)      # (Ny,1)
    cz = numpy.take(coords[2], 0, axis=0)      # (1,Nzh)

    # Hermitian weights for an axis-0 slab: 2 on modes with positive
    # compressed-axis frequency, else 1 (meshtools.py:188-215);
    # everything weight-1 for a non-compressed (real) field
    if hermitian:
        nonsingular = numpy.broadcast_to(cz > 0.,
                                         (cy.shape[0], cz.shape[1]))
    else:
        nonsingular = numpy.zeros((cy.shape[0], cz.shape[1]), dtype=bool)
    hw = numpy.ones(nonsingular.shape, dtype='f8')
    hw[nonsingular] = 2.

    for islab in range(y3d.shape[0]):
        cx = float(coords[0][islab, 0, 0])

        # cx*cx, not cx**2: python's scalar pow rounds differently from
        # the product in the last ulp for a handful of values (observed
        # at Nmesh=512, Box=2500), while the reference's norm2 is a
        # numpy ARRAY power (meshtools.py:117), which is elementwise
        # x*x — modes sitting exactly on a k-bin edge digitize by this
        # ulp (the GPU kernel also computes kx*kx)
        xslab = cx * cx + cy * cy + cz * cz           # norm2 (Ny,Nzh)
        if xslab.size == 0:
            continue

        dig_x = numpy.digitize(xslab.ravel(), x2edges)

        norm = xslab ** 0.5
        with numpy.errstate(invalid='ignore', divide='ignore'):
            mu = (cx * los[0] + cy * los[1] + cz * los[2]) / norm
        mu = numpy.where(norm == 0.0, 0.0, mu)
        dig_mu = numpy.digitize(mu.ravel(), muedges)

        multi_index = numpy.ravel_multi_index([dig_x, dig_mu],
                                              (Nx + 2, Nmu + 2))

        xsum.flat += numpy.bincount(multi_index,
                                    weights=(norm * hw).ravel(),
                                    minlength=xsum.size)
        Nsum.flat += numpy.bincount(multi_index, weights=hw.ravel(),
                                    minlength=Nsum.size).astype('i8')

        for iell, ell in enumerate(_poles):
            weighted = legpoly[iell](mu) * y3d[islab]
            # conjugate-pair identity on the doubled modes (:649-656)
            if not hermitian:
                pass
            elif ell % 2:
                weighted = 1j * numpy.where(nonsingular,
                                            2. * weighted.imag,
                                            weighted.imag) \
                    + numpy.where(nonsingular, 0., weighted.real)
            else:
                weighted = numpy.where(nonsingular, 2. * weighted.real,
                                       weighted.real) \
                    + 1j * numpy.where(nonsingular, 0., weighted.imag)
            weighted = weighted * (2. * ell + 1.)
            ysum[iell].real.flat += numpy.bincount(
                multi_index, weights=weighted.real.ravel(),
                minlength=Nsum.size)
            if numpy.iscomplexobj(ysum):
                ysum[iell].imag.flat += numpy.bincount(
                    multi_index, weights=weighted.imag.ravel(),
                    minlength=Nsum.size)

        musum.flat += numpy.bincount(multi_index,
                                     weights=(mu * hw).ravel(),
                                     minlength=musum.size)

    if _return_sums:
        # raw per-slab-range accumulators (pre-fold): lets callers split
        # the field along axis 0 across workers and add the partials
        # (bench.py's 8-core cpu_baseline harness); sums are linear so
        # the composition is exact
        return (xsum, musum, ysum, Nsum)

    # fold internal mu==1 bin into the last visible bin (:674-679)
    ysum[..., -2] += ysum[..., -1]
    musum[:, -2] += musum[:, -1]
    xsum[:, -2] += xsum[:, -1]
    Nsum[:, -2] += Nsum[:, -1]

    sl = slice(1, -1)
    with numpy.errstate(invalid='ignore', divide='ignore'):
        y2d = (ysum[0] / Nsum)[sl, sl]
        xmean_2d = (xsum / Nsum)[sl, sl]
        mumean_2d = (musum / Nsum)[sl, sl]
        N_2d = Nsum[sl, sl]

        if do_poles:
            N_1d = Nsum[sl, sl].sum(axis=-1)
            xmean_1d = xsum[sl, sl].sum(axis=-1) / N_1d
            pole_arr = ysum[:, sl, sl].sum(axis=-1) / N_1d
            pole_arr = pole_arr[ell_idx, ...]
            pole_result = (xmean_1d, pole_arr, N_1d)
        else:
            pole_result = None

    return (xmean_2d, mumean_2d, y2d, N_2d), pole_result


# ---------------------------------------------------------------- driver

def fftpower_oracle(position, Nmesh, BoxSize, mode='1d', second_position=None,
                    weight=None, second_weight=None, resampler='cic',
                    compensated=True, interlaced=False, los=(0, 0, 1),
                    Nmu=5, dk=None, kmin=0., kmax=None, poles=(),
                    paint_chunk_size=4 * 1024 * 1024):
    """
    End-to-end FFTPower on numpy: the CPU oracle.  Mirrors
    FFTPower.__init__/run (fftpower.py:194-334) with the FFTPower cast
    defaults dtype='f8', compensated=True (fftpower.py:717).

    Returns a dict: kedges, muedges, k, mu, power (complex), modes,
    poles (dict ell->array) or None, pole_k, pole_modes, attrs.
    """
    geom = MeshGeometry(Nmesh, BoxSize, dtype='f8')

    if mode not in ('1d', '2d'):
        raise ValueError("`mode` should be either '1d' or '2d'")
    if mode == '1d':
        Nmu = 1

    def make_complex(pos, wgt):
        mesh, attrs = to_real_field(pos, geom, weight=wgt,
                                    resampler=resampler,
                                    interlaced=interlaced,
                                    paint_chunk_size=paint_chunk_size)
        c = r2c(mesh, geom)
        if compensated:
            apply_compensation(c, geom, resampler, interlaced)
        return c, attrs

    c1, attrs1 = make_complex(position, weight)
    auto = second_position is None
    if auto:
        c2, attrs2 = c1, attrs1
    else:
        c2, attrs2 = make_complex(second_position, second_weight)

    p3d = compute_3d_power(c1, c2, geom)

    if dk is None:
        dk = 2 * numpy.pi / geom.BoxSize.min()
    if kmax is None:
        kmax_used = numpy.pi * geom.Nmesh.min() / geom.BoxSize.max() + dk / 2
    else:
        kmax_used = kmax

    kcoords = None
    if dk > 0:
        kedges = numpy.arange(kmin, kmax_used, dk)
    else:
        kedges, kcoords = kedges_unique(geom, kmax_used)

    muedges = numpy.linspace(-1, 1, Nmu + 1, endpoint=True)

    result, pole_result = project_to_basis(p3d, geom, [kedges, muedges],
                                           los=los, poles=poles)
    xmean_2d, mumean_2d, y2d, N_2d = result

    attrs = {
        'N1': attrs1['N'], 'N2': attrs2['N'],
        'shotnoise': attrs1['shotnoise'] if auto else 0.0,
        'mode': mode, 'los': list(los), 'Nmu': Nmu, 'poles': list(poles),
        'dk': dk, 'kmin': kmin, 'kmax': kmax,
        'Nmesh': geom.Nmesh.copy(), 'BoxSize': geom.BoxSize.copy(),
        'volume': float(numpy.prod(geom.BoxSize)),
    }

    out = {
        'kedges': kedges, 'muedges': muedges, 'kcoords': kcoords,
        'k': numpy.squeeze(xmean_2d) if mode == '1d' else xmean_2d,
        'mu': None if mode == '1d' else mumean_2d,
        'power': numpy.squeeze(y2d) if mode == '1d' else y2d,
        'modes': numpy.squeeze(N_2d) if mode == '1d' else N_2d,
        'attrs': attrs,
    }
    if pole_result is not None:
        pole_k, pole_arr, pole_modes = pole_result
        out['pole_k'] = pole_k
        out['poles'] = {ell: pole_arr[i] for i, ell in enumerate(poles)}
        out['pole_modes'] = pole_modes
    else:
        out['poles'] = None
    return out


def fftcorr_oracle(position, Nmesh, BoxSize, mode='1d',
                   second_position=None, weight=None, second_weight=None,
                   resampler='cic', compensated=True, interlaced=False,
                   los=(0, 0, 1), Nmu=5, dr=None, rmin=0., rmax=None,
                   poles=(), paint_chunk_size=4 * 1024 * 1024):
    """
    End-to-end FFTCorr oracle (reference algorithms/fftcorr.py:15-235):
    the 3D power transformed back with c2r and divided by V (:151-158),
    binned in configuration space with mu in [0, 1] (:175) and
    dr defaulting to BoxSize.min()/Nmesh.max() (:89).
    """
    geom = MeshGeometry(Nmesh, BoxSize, dtype='f8')
    if mode not in ('1d', '2d'):
        raise ValueError("`mode` should be either '1d' or '2d'")
    if mode == '1d':
        Nmu = 1

    def make_complex(pos, wgt):
        mesh, attrs = to_real_field(pos, geom, weight=wgt,
                                    resampler=resampler,
                                    interlaced=interlaced,
                                    paint_chunk_size=paint_chunk_size)
        c = r2c(mesh, geom)
        if compensated:
            apply_compensation(c, geom, resampler, interlaced)
        return c, attrs

    c1, attrs1 = make_complex(position, weight)
    auto = second_position is None
    if auto:
        c2f, attrs2 = c1, attrs1
    else:
        c2f, attrs2 = make_complex(second_position, second_weight)

    p3d = compute_3d_power(c1, c2f, geom)
    y3d = c2r(p3d, geom)
    y3d *= 1.0 / float(numpy.prod(geom.BoxSize))   # fftcorr.py:157-158

    if dr is None:
        dr = float(geom.BoxSize.min() / geom.Nmesh.max())
    if rmax is None:
        rmax = 0.5 * float(geom.BoxSize.min()) + dr / 2
    rcoords = None
    if dr > 0:
        redges = numpy.arange(rmin, rmax, dr)
    else:
        redges, rcoords = redges_unique(geom, rmax)
    muedges = numpy.linspace(0, 1, Nmu + 1, endpoint=True)

    result, pole_result = project_to_basis(
        y3d, geom, [redges, muedges], los=los, poles=poles,
        coords=real_coords(geom), hermitian=False)
    xmean_2d, mumean_2d, y2d, N_2d = result

    attrs = {
        'N1': attrs1['N'], 'N2': attrs2['N'],
        'shotnoise': attrs1['shotnoise'] if auto else 0.0,
        'mode': mode, 'los': list(los), 'Nmu': Nmu, 'poles': list(poles),
        'dr': dr, 'rmin': rmin, 'rmax': rmax,
        'Nmesh': geom.Nmesh.copy(), 'BoxSize': geom.BoxSize.copy(),
    }
    out = {
        'redges': redges, 'muedges': muedges, 'rcoords': rcoords,
        'r': numpy.squeeze(xmean_2d) if mode == '1d' else xmean_2d,
        'mu': None if mode == '1d' else mumean_2d,
        'corr': numpy.squeeze(y2d) if mode == '1d' else y2d,
        'modes': numpy.squeeze(N_2d) if mode == '1d' else N_2d,
        'attrs': attrs,
    }
    if pole_result is not None:
        pole_k, pole_arr, pole_modes = pole_result
        out['pole_r'] = pole_k
        out['poles'] = {ell: pole_arr[i] for i, ell in enumerate(poles)}
        out['pole_modes'] = pole_modes
    else:
        out['poles'] = None
    return out
