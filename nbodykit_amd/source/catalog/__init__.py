from .uniform import RandomCatalog, UniformCatalog
from .lognormal import LogNormalCatalog
from .array import ArrayCatalog
