"""
Communicator abstraction replacing mpi4py.

The reference uses mpi4py communicators everywhere; the hot-path
collectives are: ``allreduce`` of small binning arrays and scalars
(nbodykit/algorithms/fftpower.py:669-672), ``allgather`` of small vectors
(fftpower.py:755, source/mesh/catalog.py:222), ``bcast`` (fftpower.py:84)
and ``alltoall`` of array payloads (utils.py:367-369, pmesh exchange).

Here a :class:`Comm` is either serial (size 1, no backend) or backed by an
initialized ``torch.distributed`` process group — gloo on CPU for tests,
nccl (= RCCL over xGMI on ROCm) for the 1-process-per-GPU runs.  Small
object collectives use the *_object torch APIs; numeric array reductions go
through tensors on the backend's device.
"""
import numpy


class Comm(object):
    """Base interface; also the serial (size-1) implementation."""

    rank = 0
    size = 1

    def allgather(self, obj):
        return [obj]

    def allreduce(self, value, op='sum'):
        # identity for size-1; keep numpy semantics (arrays pass through)
        return value

    def bcast(self, obj, root=0):
        return obj

    def barrier(self):
        pass

    def alltoall(self, objs):
        assert len(objs) == self.size
        return list(objs)

    def __repr__(self):
        return "<%s rank=%d size=%d>" % (type(self).__name__, self.rank, self.size)


class SerialComm(Comm):
    pass


class TorchComm(Comm):
    """A Comm over an initialized torch.distributed process group."""

    def __init__(self, group=None):
        import torch.distributed as dist
        self._dist = dist
        self._group = group
        self.rank = dist.get_rank(group)
        self.size = dist.get_world_size(group)
        backend = dist.get_backend(group)
        self._device = 'cuda' if str(backend) == 'nccl' else 'cpu'

    def allgather(self, obj):
        out = [None] * self.size
        self._dist.all_gather_object(out, obj, group=self._group)
        return out

    def allreduce(self, value, op='sum'):
        import torch
        arr = numpy.asarray(value)
        t = torch.from_numpy(numpy.ascontiguousarray(arr)).to(self._device)
        ops = {'sum': self._dist.ReduceOp.SUM,
               'min': self._dist.ReduceOp.MIN,
               'max': self._dist.ReduceOp.MAX}
        self._dist.all_reduce(t, op=ops[op], group=self._group)
        result = t.cpu().numpy()
        if numpy.isscalar(value) or numpy.ndim(value) == 0:
            return result.item() if hasattr(result, 'item') else result
        return result.reshape(arr.shape)

    def bcast(self, obj, root=0):
        box = [obj if self.rank == root else None]
        self._dist.broadcast_object_list(box, src=root, group=self._group)
        return box[0]

    def barrier(self):
        self._dist.barrier(group=self._group)

    def alltoall(self, objs):
        # object alltoall built from size gathers (payloads here are small:
        # RNG front-padding slices, plan metadata). Bulk particle exchange
        # uses tensor all_to_all_single in the GPU layer, not this.
        assert len(objs) == self.size
        gathered = self.allgather(objs)
        return [gathered[src][self.rank] for src in range(self.size)]


def default_comm():
    """A TorchComm if torch.distributed is initialized, else serial."""
    try:
        import torch.distributed as dist
        if dist.is_available() and dist.is_initialized():
            return TorchComm()
    except ImportError:
        pass
    return SerialComm()
