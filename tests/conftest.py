import os
import sys

import numpy
import pytest

# repo root on sys.path so `nbodykit_amd` and `oracle` import in-tree
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU and the HIP extension")


@pytest.fixture
def serial_comm():
    from nbodykit_amd.comm import SerialComm
    return SerialComm()


def uniform_positions(nbar, BoxSize, seed, comm=None):
    """The UniformCatalog position recipe (reference
    source/catalog/uniform.py:94-100): serial Poisson for N, then
    MPIRandomState uniforms scaled by the box."""
    from nbodykit_amd.comm import SerialComm
    from nbodykit_amd.mpirng import MPIRandomState
    if comm is None:
        comm = SerialComm()
    N = numpy.random.RandomState(seed).poisson(nbar * BoxSize ** 3)
    start = comm.rank * N // comm.size
    end = (comm.rank + 1) * N // comm.size
    rng = MPIRandomState(comm, seed=seed, size=end - start)
    return rng.uniform(itemshape=(3,)) * BoxSize
