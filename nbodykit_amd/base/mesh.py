"""
MeshSource — the public drop-in boundary (reference
nbodykit/base/mesh.py:6-433): ``compute(mode, Nmesh)`` (:246-250), the
deprecated ``paint()`` alias (:252-254), the action pipeline with
automatic r2c/c2r casts (:256-338), ``apply(func, kind, mode)`` views
(:118-176), ``preview`` and JSON-able ``attrs``.  Fields underneath are
the GPU-backed ``nbodykit_amd.pm`` containers.
"""
import logging
import warnings

import numpy

from nbodykit_amd.pm import ParticleMesh, RealField, ComplexField


class MeshSource(object):
    logger = logging.getLogger('MeshSource')

    def __init__(self, comm, Nmesh, BoxSize, dtype):
        self.comm = comm
        self.dtype = dtype

        if Nmesh is None or BoxSize is None:
            raise ValueError("both Nmesh and BoxSize must not be None to "
                             "initialize ParticleMesh")

        self.pm = ParticleMesh(BoxSize=BoxSize, Nmesh=Nmesh,
                               dtype=dtype, comm=self.comm)
        self.attrs['BoxSize'] = self.pm.BoxSize.copy()
        self.attrs['Nmesh'] = self.pm.Nmesh.copy()

        self._actions = []
        self.base = None

    def __finalize__(self, other):
        if isinstance(other, MeshSource):
            self.comm = other.comm
            self.dtype = other.dtype
            self.pm = other.pm
            self.attrs.update(other.attrs)
            self._actions = []
            self._actions.extend(other.actions)
        return self

    def view(self):
        view = object.__new__(MeshSource)
        view.base = self
        return view.__finalize__(self)

    @property
    def attrs(self):
        try:
            return self._attrs
        except AttributeError:
            self._attrs = {}
            return self._attrs

    @property
    def actions(self):
        return self._actions

    def apply(self, func, kind='wavenumber', mode='complex'):
        """Append a (mode, func, kind) action and return a view
        (reference :118-176)."""
        if isinstance(func, type) and issubclass(func, MeshFilter):
            func = func()
        if isinstance(func, MeshFilter):
            mode = func.mode
            kind = func.kind
            func = func.filter

        assert mode in ('complex', 'real'), \
            "``mode`` should be 'complex' or 'real'"
        if mode == 'real':
            assert kind in ('relative', 'index')
        else:
            assert kind in ('wavenumber', 'circular', 'index')
        view = self.view()
        view._actions.append((mode, func, kind))
        return view

    def __len__(self):
        return 0

    def to_real_field(self, out=None, normalize=True):
        if isinstance(self.base, MeshSource):
            return self.base.to_real_field(out=out, normalize=normalize)
        return NotImplemented

    def to_complex_field(self, out=None):
        if isinstance(self.base, MeshSource):
            return self.base.to_complex_field(out=out)
        return NotImplemented

    def to_field(self, mode='real', out=None):
        if mode == 'real':
            real = self.to_real_field()
            if real is NotImplemented:
                cplx = self.to_complex_field()
                assert cplx is not NotImplemented
                real = cplx.c2r(out=Ellipsis)
                if hasattr(cplx, 'attrs'):
                    real.attrs = cplx.attrs
            var = real
        elif mode == 'complex':
            cplx = self.to_complex_field()
            if cplx is NotImplemented:
                real = self.to_real_field()
                assert real is not NotImplemented
                cplx = real.r2c(out=Ellipsis)
                if hasattr(real, 'attrs'):
                    cplx.attrs = real.attrs
            var = cplx
        else:
            raise ValueError("mode is either real or complex, %s given"
                             % mode)
        return var

    def compute(self, mode='real', Nmesh=None):
        """Compute the mesh into a RealField or ComplexField, applying
        the action pipeline (reference :246-250)."""
        return self._paint_XXX(mode=mode, Nmesh=Nmesh)

    def paint(self, mode='real', Nmesh=None):
        warnings.warn("the paint method is deprecated from the Public "
                      "API. Use .compute() instead.", DeprecationWarning)
        return self._paint_XXX(mode=mode, Nmesh=Nmesh)

    def _paint_XXX(self, mode='real', Nmesh=None):
        if mode not in ('real', 'complex'):
            raise ValueError('mode must be "real" or "complex"')

        actions = self.actions + [(mode,)]

        # start in the mode of the first action to skip useless casts
        var = self.to_field(mode=actions[0][0])

        attrs = var.attrs if hasattr(var, 'attrs') else {}

        for action in actions:
            if action[0] == 'complex' and not isinstance(var, ComplexField):
                var = var.r2c(out=Ellipsis)
            if action[0] == 'real' and not isinstance(var, RealField):
                var = var.c2r(out=Ellipsis)

            if len(action) > 1:
                kwargs = {'func': action[1], 'out': Ellipsis}
                if action[2] is not None:
                    kwargs['kind'] = action[2]
                var.apply(**kwargs)

        var = var.cast() if isinstance(var, ComplexField) else var

        if Nmesh is not None and any(
                numpy.asarray(Nmesh).ravel() != self.pm.Nmesh):
            # resample by copying overlapping Fourier modes
            # (reference base/mesh.py:320-330)
            from nbodykit_amd.pm import ParticleMesh, spectral_resample
            new_pm = ParticleMesh(BoxSize=self.pm.BoxSize, Nmesh=Nmesh,
                                  dtype=self.dtype, comm=self.comm)
            want_real = isinstance(var, RealField)
            if want_real:
                var = var.r2c(out=Ellipsis)
            var = spectral_resample(var, new_pm)
            if want_real:
                var = var.c2r(out=Ellipsis)
            if self.comm.rank == 0:
                self.logger.info('%s resampling from %s to %s done'
                                 % (str(self), str(self.pm.Nmesh),
                                    str(new_pm.Nmesh)))

        var.attrs = attrs
        var.attrs.update(self.attrs)
        return var

    def preview(self, axes=None, Nmesh=None, root=0):
        field = self.to_field(mode='real')
        if Nmesh is None:
            Nmesh = self.pm.Nmesh
        return field.preview(Nmesh, axes=axes)

    def save(self, output, dataset='Field', mode='real'):
        """Save the mesh as a bigfile readable by
        :class:`~nbodykit_amd.source.mesh.bigfile.BigFileMesh`
        (reference base/mesh.py:444-500): the raveled local field rows
        ordered by rank, with ``ndarray.shape``/``BoxSize``/``Nmesh``
        attrs and the remaining attrs JSON-encoded."""
        import json
        import warnings
        from nbodykit_amd.io.bigfile_format import BigFile
        from nbodykit_amd.utils import JSONEncoder

        field = self.compute(mode=mode)
        data = field.value.cpu().numpy().ravel()

        with BigFile(output, create=True, comm=self.comm) as ff:
            bb = ff.create_from_array(dataset, data)
            cshape = [int(n) for n in self.pm.Nmesh]
            if mode == 'complex':
                cshape[-1] = cshape[-1] // 2 + 1
            bb.attrs['ndarray.shape'] = numpy.asarray(cshape, dtype='i8')
            bb.attrs['BoxSize'] = self.pm.BoxSize
            bb.attrs['Nmesh'] = self.pm.Nmesh
            for key in field.attrs:
                if key in bb.attrs:
                    continue
                value = field.attrs[key]
                try:
                    bb.attrs[key] = value
                except (ValueError, TypeError):
                    try:
                        bb.attrs[key] = 'json://' + json.dumps(
                            value, cls=JSONEncoder)
                    except Exception:
                        warnings.warn(
                            "attribute %s of type %s is unsupported and "
                            "lost while saving MeshSource"
                            % (key, type(value)))


class MeshFilter(object):
    """Base class for named filters (reference base/mesh.py:414-433)."""
    kind = None
    mode = None

    def filter(self, x, v):
        raise NotImplementedError
