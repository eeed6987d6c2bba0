from .catalog import CatalogMesh
from .field import FieldMesh
