"""
nbodykit_amd — an MI355X-native rebuild of nbodykit's FFTPower hot path.

The public API mirrors nbodykit (bccp/nbodykit v0.3.16):
``CatalogSource.to_mesh()`` / ``MeshSource.compute()`` / ``FFTPower`` —
see ``nbodykit/base/mesh.py:246-338`` and ``nbodykit/algorithms/fftpower.py``
in the reference.  The compute underneath is hand-written HIP (gfx950)
driven through a ctypes C-ABI (``include/nbk_hip.h``), with RCCL over xGMI
(via ``torch.distributed``) replacing mpi4py.

This module provides the communicator stack (``CurrentMPIComm``,
mirroring ``nbodykit/__init__.py:107-191``), the global options dict with
``set_options`` (``nbodykit/__init__.py:22-25,215-256``) and
``setup_logging`` (``nbodykit/__init__.py:259-300``).
"""
import logging
import time
from contextlib import contextmanager

__version__ = "0.1.0"

# Same three option names as the reference (nbodykit/__init__.py:22-25).
# paint_chunk_size is the GPU paint batch size analogue.
_global_options = {
    # the reference defaults to 4M particles per paint chunk (a CPU
    # memory-safety choice); with 288 GB of HBM a 2^28 chunk fits
    # comfortably and keeps the two-level sort + gather paint engaged
    # for big catalogs (set_options(paint_chunk_size=...) restores any
    # other value)
    'paint_chunk_size': 1 << 28,
    'dask_chunk_size': 100000,
    'global_cache_size': 1e8,
    # paint locality-sort thresholds (tests shrink these to exercise the
    # two-level atomic-free sort + gather paint on small inputs)
    'sort_min_n': 1 << 21,
    'sort_two_level_min_n': 1 << 22,
    'sort_two_level_min_cells': 1 << 23,
}


class set_options(object):
    """Context manager to temporarily override ``_global_options``
    (reference: nbodykit/__init__.py:215-256)."""

    def __init__(self, **kwargs):
        self.old = _global_options.copy()
        for key in kwargs:
            if key not in _global_options:
                raise KeyError("unknown global option '%s'" % key)
        _global_options.update(kwargs)

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        _global_options.clear()
        _global_options.update(self.old)


class CurrentMPIComm(object):
    """
    A stack of communicators, mirroring ``nbodykit.CurrentMPIComm``
    (nbodykit/__init__.py:107-191).  Communicators here are
    :class:`nbodykit_amd.comm.Comm` objects (serial, or torch.distributed
    backed) instead of mpi4py communicators.
    """
    _stack = None
    logger = logging.getLogger("CurrentMPIComm")

    @staticmethod
    def enable(func):
        """Decorator that injects ``comm=CurrentMPIComm.get()`` when the
        caller passed ``comm=None`` (reference :128-147)."""
        import functools

        @functools.wraps(func)
        def wrapped(*args, **kwargs):
            kwargs.setdefault('comm', None)
            if kwargs['comm'] is None:
                kwargs['comm'] = CurrentMPIComm.get()
            return func(*args, **kwargs)
        return wrapped

    @classmethod
    def _init_stack(cls):
        if cls._stack is None:
            from nbodykit_amd.comm import default_comm
            cls._stack = [default_comm()]

    @classmethod
    def get(cls):
        """The communicator on top of the stack (reference :171-176)."""
        cls._init_stack()
        return cls._stack[-1]

    @classmethod
    def push(cls, comm):
        cls._init_stack()
        cls._stack.append(comm)

    @classmethod
    def pop(cls):
        cls._init_stack()
        if len(cls._stack) == 1:
            raise RuntimeError("cannot pop the last communicator")
        return cls._stack.pop()

    @classmethod
    @contextmanager
    def enter(cls, comm):
        cls.push(comm)
        try:
            yield
        finally:
            cls.pop()


_logging_handler = None


def setup_logging(log_level="info"):
    """
    Per-rank prefixed logging, matching the reference's
    ``[ elapsed ] rank:`` format (nbodykit/__init__.py:259-300).
    """
    levels = {
        "info": logging.INFO,
        "debug": logging.DEBUG,
        "warning": logging.WARNING,
        "error": logging.ERROR,
    }
    import sys
    t0 = time.time()

    comm = CurrentMPIComm.get()
    rank = comm.rank

    class Formatter(logging.Formatter):
        def format(self, record):
            s1 = ('[ %09.2f ] % 3d: ' % (time.time() - t0, rank))
            return s1 + logging.Formatter.format(self, record)

    fmt = Formatter(fmt='%(asctime)s %(name)-15s %(levelname)-8s %(message)s',
                    datefmt='%m-%d %H:%M ')

    global _logging_handler
    if _logging_handler is None:
        _logging_handler = logging.StreamHandler(sys.stdout)
        logging.getLogger().addHandler(_logging_handler)

    _logging_handler.setFormatter(fmt)
    logging.getLogger().setLevel(levels[log_level])
