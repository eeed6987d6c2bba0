"""
FieldMesh — wrap an existing Real/ComplexField as a MeshSource
(reference nbodykit/source/mesh/field.py; used by FFTPower's
_cast_source, algorithms/fftpower.py:711-713).
"""
from nbodykit_amd.base.mesh import MeshSource
from nbodykit_amd.pm import RealField, ComplexField


class FieldMesh(MeshSource):

    def __init__(self, field):
        if not isinstance(field, (RealField, ComplexField)):
            raise TypeError("FieldMesh takes a RealField or ComplexField")
        self.field = field
        MeshSource.__init__(self, field.pm.comm, field.pm.Nmesh,
                            field.pm.BoxSize, 'f8')
        self.attrs.update(field.attrs)

    def to_real_field(self, out=None, normalize=True):
        if isinstance(self.field, RealField):
            return self.field.copy()
        return NotImplemented

    def to_complex_field(self, out=None):
        if isinstance(self.field, ComplexField):
            return self.field.copy()
        return NotImplemented
