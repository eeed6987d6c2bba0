from .catalog import FKPCatalog, FKPWeightFromNbar
from .catalogmesh import FKPCatalogMesh
from .fkp import ConvolvedFFTPower, get_real_Ylm
