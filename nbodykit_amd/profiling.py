"""
Lightweight HIP-event timing of named kernel regions, recorded on the
stream the kernels are launched on (torch's current stream — the same
one every nbk_* call receives).  Used by bench.py for the roofline
numbers; disabled (zero overhead) unless enabled.
"""
from contextlib import contextmanager

_enabled = False
_records = {}        # name -> list of (start_event, end_event, units)


def enable():
    global _enabled
    _enabled = True
    _records.clear()


def disable():
    global _enabled
    _enabled = False


def reset():
    _records.clear()


@contextmanager
def collect(name, units=0):
    """Bracket a launch region with HIP events when enabled.  ``units``
    is the work count of the region (e.g. particle deposits issued)."""
    if not _enabled:
        yield
        return
    import torch
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    start.record()
    yield
    end.record()
    _records.setdefault(name, []).append((start, end, units))


def summary():
    """{name: {'ms': total, 'calls': n, 'units': total}} — synchronizes."""
    import torch
    torch.cuda.synchronize()
    out = {}
    for name, recs in _records.items():
        ms = sum(s.elapsed_time(e) for s, e, _ in recs)
        units = sum(u for _, _, u in recs)
        out[name] = {'ms': ms, 'calls': len(recs), 'units': units}
    return out
