"""
Utilities on the hot path: FrontPadArray (needed by MPIRandomState;
reference nbodykit/utils.py:350-370) and the JSON encoder/decoder pair used
by FFTPower.save/load and BinnedStatistic.to_json (reference
nbodykit/utils.py:381-489 — same ``__dtype__/__shape__/__data__`` and
``__complex__`` schema, so files round-trip with real nbodykit).
"""
import json
import numpy


def FrontPadArray(array, front, comm):
    """
    Pad ``array`` in front with the last ``front`` items collectively owned
    by earlier ranks (reference nbodykit/utils.py:350-370).

    Serial case (size 1): ``front`` must be 0 since there is no earlier rank.
    """
    N = numpy.array(comm.allgather(len(array)), dtype='intp')
    offsets = numpy.concatenate([[0], numpy.cumsum(N)])
    mystart = offsets[comm.rank] - front

    # how many items to receive from each earlier rank: the window
    # [mystart, mystart+front) ends exactly at this rank's start, so only
    # earlier ranks contribute, each with its tail items.  Ranks whose end
    # lies beyond the window (self and later) are zeroed, not clamped.
    torecv = (offsets[:-1] + N) - mystart
    torecv[torecv < 0] = 0          # entirely before the window
    torecv[torecv > front] = 0      # self / later ranks: beyond the window
    over = torecv > N
    torecv[over] = N[over]          # window fully encloses that rank

    if comm.allreduce(int(torecv.sum() != front), op='sum') != 0:
        raise ValueError("cannot plan front-padding: requested %d items but "
                         "only %d available before this rank"
                         % (front, torecv.sum()))

    tosend = comm.alltoall(list(torecv))
    sendbuf = [array[-items:] if items > 0 else array[0:0] for items in tosend]
    recvbuf = comm.alltoall(sendbuf)
    return numpy.concatenate(list(recvbuf) + [array], axis=0)


class JSONEncoder(json.JSONEncoder):
    """Encode numpy arrays / scalars / complex values like the reference
    (nbodykit/utils.py:380-430); Cosmology objects serialize their
    parameter dict under ``__cosmo__``."""

    def default(self, obj):
        from nbodykit_amd.cosmology import Cosmology
        if isinstance(obj, Cosmology):
            return {'__cosmo__': dict(obj.pars)}

        if isinstance(obj, complex):
            return {'__complex__': [obj.real, obj.imag]}

        if isinstance(obj, numpy.ndarray):
            dtype = obj.dtype
            return {
                '__dtype__': dtype.str if dtype.names is None else dtype.descr,
                '__shape__': obj.shape,
                '__data__': obj.tolist(),
            }
        if isinstance(obj, numpy.floating):
            return float(obj)
        if isinstance(obj, numpy.integer):
            return int(obj)
        if isinstance(obj, numpy.complexfloating):
            return {'__complex__': [float(obj.real), float(obj.imag)]}
        if isinstance(obj, numpy.bool_):
            return bool(obj)
        return json.JSONEncoder.default(self, obj)


class JSONDecoder(json.JSONDecoder):
    """Inverse of :class:`JSONEncoder` (reference nbodykit/utils.py:432-489)."""

    @staticmethod
    def hook(value):

        def fixdtype(dtype):
            # JSON turns dtype.descr tuples into lists; restore tuples/strs
            if isinstance(dtype, list):
                fixed = []
                for field in dtype:
                    if len(field) == 3:
                        fixed.append((str(field[0]), str(field[1]), field[2]))
                    elif len(field) == 2:
                        fixed.append((str(field[0]), str(field[1])))
                return fixed
            return dtype

        def fixdata(data, depth, dtype):
            # structured arrays need the innermost dimension as tuples
            if not isinstance(dtype, list):
                return data
            if depth > 0:
                return [fixdata(item, depth - 1, dtype) for item in data]
            assert len(data) == len(dtype)
            return tuple(data)

        if '__dtype__' in value:
            dtype = fixdtype(value['__dtype__'])
            shape = value['__shape__']
            data = fixdata(value['__data__'], len(shape), dtype)
            return numpy.array(data, dtype=dtype)

        if '__cosmo__' in value:
            from nbodykit_amd.cosmology import Cosmology
            return Cosmology(**value['__cosmo__'])

        if '__complex__' in value:
            real, imag = value['__complex__']
            return real + 1j * imag

        return value

    def __init__(self, *args, **kwargs):
        kwargs['object_hook'] = JSONDecoder.hook
        json.JSONDecoder.__init__(self, *args, **kwargs)


def timer(start, end):
    """Format an elapsed-seconds interval as h:mm:ss (reference utils.py:491)."""
    hours, rem = divmod(end - start, 3600)
    minutes, seconds = divmod(rem, 60)
    return "%d:%02d:%05.2f" % (int(hours), int(minutes), seconds)
