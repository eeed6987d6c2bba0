from .fftpower import (FFTPower, FFTBase, ProjectedFFTPower,
                       project_to_basis)
from .fftcorr import FFTCorr
from .fftrecon import FFTRecon
from .convpower import ConvolvedFFTPower, FKPCatalog, FKPWeightFromNbar
from .zhist import RedshiftHistogram
