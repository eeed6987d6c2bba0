"""
ORACLE — TEST INFRASTRUCTURE, NOT PRODUCT CODE.

Single-process numpy restatement of nbodykit's FFTPower hot path
(CatalogSource.to_mesh -> CIC/TSC/PCS paint -> R2C FFT -> |delta(k)|^2
k/mu binning -> P(k)), used ONLY as the parity checker for the HIP/GPU
product path.  Only ``tests/``, ``__graft_entry__.smoke()`` and
``bench.py``'s ``cpu_baseline`` leg may import this package; the product
(``nbodykit_amd``) never does, and fails loudly when its HIP extension or
a GPU is missing.

Each function cites the reference (bccp/nbodykit v0.3.16) file:line it
restates.  Parity pinning status: the reference's own property tests are
ported in ``tests/test_oracle_fftpower.py`` (flat shot noise of a
compensated uniform paint, poles/monopole identity, unique edges,
zero-mode clear, Hermitian weights, chunk invariance, shotnoise value);
exact-value P(k) parity vs *upstream* pmesh+pfft is UNPINNED because the
reference repo contains no FFTPower golden vectors and pmesh/pfft/mpi4py
cannot be imported or built in this environment.
"""
from .mesh import MeshGeometry, r2c, c2r, complex_coords, real_coords
from .paint import paint, readout, WINDOW_SUPPORT
from .catalogmesh import to_real_field
from .fftpower import (compensation_filter, apply_compensation,
                      compute_3d_power, project_to_basis,
                      fftpower_oracle, fftcorr_oracle)
from .fftrecon import fftrecon_oracle
from .convpower import convpower_oracle, real_Ylm
