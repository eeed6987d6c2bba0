from .catalog import CatalogMesh
from .field import FieldMesh
from .array import ArrayMesh
from .linear import LinearMesh
from .bigfile import BigFileMesh
