"""
The user-facing namespace (reference nbodykit/lab.py): the hot-path
algorithm, sources and cosmology in one import:

    from nbodykit_amd.lab import *
"""
import numpy

from nbodykit_amd import CurrentMPIComm, setup_logging, set_options
from nbodykit_amd import cosmology
from nbodykit_amd.cosmology import Cosmology, Planck15, LinearPower
from nbodykit_amd.algorithms import (FFTPower, FFTCorr, FFTRecon,
                                     ProjectedFFTPower, project_to_basis)
from nbodykit_amd.source.catalog import (UniformCatalog, RandomCatalog,
                                         LogNormalCatalog, ArrayCatalog)
from nbodykit_amd.source.mesh import (CatalogMesh, FieldMesh,
                                      ArrayMesh, LinearMesh)
from nbodykit_amd.source.mesh.bigfile import BigFileMesh
from nbodykit_amd.source.catalog.bigfile import BigFileCatalog
from nbodykit_amd.source.catalog.file import (BinaryCatalog,
    CSVCatalog, Gadget1Catalog)
from nbodykit_amd.base.catalog import CatalogSource
from nbodykit_amd.base.mesh import MeshSource
from nbodykit_amd.binned_statistic import BinnedStatistic
from nbodykit_amd.pm import ParticleMesh, RealField, ComplexField
from nbodykit_amd import transform
from nbodykit_amd.algorithms.convpower import (ConvolvedFFTPower,
                                               FKPCatalog,
                                               FKPWeightFromNbar)
from nbodykit_amd.algorithms.zhist import RedshiftHistogram
from nbodykit_amd import filters
from nbodykit_amd.source.catalog.species import MultipleSpeciesCatalog
from nbodykit_amd import io as IO
