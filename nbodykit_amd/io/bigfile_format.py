"""
Pure-Python restatement of the **bigfile** on-disk format (the reference
delegates to the external ``bigfile`` package — rainwoodman/bigfile,
listed unpinned in reference requirements.txt:10 — which is NOT present
in this container and is not vendored under /root/reference, so the
format is restated here from its published layout).

A BigFile is a directory tree.  A *block* is a sub-directory holding

- ``header``   — text: ``DTYPE:`` (numpy dtype string, e.g. ``<f8``),
  ``NMEMB:`` (columns per row), ``NFILE:`` (number of physical files),
  then one line per physical file: ``%06X: <nrow> : <sysv checksum>``.
- ``000000`` … — the raw binary rows (little-endian per DTYPE), split
  across NFILE files.
- ``attr-v2``  — one line per attribute:
  ``<name> <dtype> <nmemb> <hex bytes> #HUMANE [ <repr> ]``.

A block named ``<ds>/.`` stores only attributes for the dataset
directory ``<ds>`` (its ``header`` has NFILE 0) — the reference relies
on this for dataset-level headers (io/bigfile.py:95-118, and
io/tests/test_bigfile.py:30-37 creates ``1/.``).

Parity note (oracle/DESIGN discipline): byte-level identity with files
written by the real C library is UNPINNED in this container (no bigfile
import, and the reference ships no committed bigfile fixtures — its io
tests generate files at run time).  What IS pinned, by the reference's
call sites and tests, is everything above the bytes: the API subset
(``File/FileMPI``, ``create``, ``create_from_array``, ``blocks``,
``Block.attrs/size/dtype``, slice reads, offset writes, nested
datasets), dtype casting on write, attrs round-tripping (including the
``json://`` convention), and the 32Mi-rows-per-file split used by
``CatalogSource.save`` (base/catalog.py:562-650).
"""
import os

import numpy


def _sysv_checksum(data):
    """System-V ``sum`` of the raw bytes (the checksum the published
    format records per physical file; not verified on read)."""
    s = int(numpy.frombuffer(data, dtype=numpy.uint8)
            .sum(dtype=numpy.uint64)) if len(data) else 0
    s = (s & 0xffff) + ((s & 0xffffffff) >> 16)
    s = (s & 0xffff) + (s >> 16)
    return s & 0xffffffff


def _normalize(name):
    """Collapse '.' path components: block '1/.' lives in directory '1'."""
    name = name.strip('/')
    parts = [p for p in name.split('/') if p not in ('', '.')]
    return '/'.join(parts)


# ------------------------------------------------------------------ attrs
def _encode_attr(value):
    """-> (dtype_str, nmemb, raw_bytes). Strings become S1 arrays (the
    published convention; how 'json://...' survives a round trip)."""
    if isinstance(value, str):
        raw = value.encode('utf-8')
        return '<S1', len(raw), raw
    arr = numpy.atleast_1d(numpy.asarray(value))
    if arr.dtype.kind in 'OU':
        raise ValueError("attribute of type %s is not storable; use the "
                         "'json://' convention" % arr.dtype)
    if arr.dtype.kind == 'b':
        arr = arr.astype('i8')
    arr = arr.astype(arr.dtype.newbyteorder('<'))
    return arr.dtype.str, arr.size, arr.tobytes()


def _decode_attr(dtype_str, nmemb, raw):
    if dtype_str.endswith('S1') or dtype_str[1] == 'S':
        return raw.decode('utf-8')
    arr = numpy.frombuffer(raw, dtype=numpy.dtype(dtype_str),
                           count=nmemb).copy()
    return arr


class AttrSet(dict):
    """Block attributes, written through to ``attr-v2`` on every set."""

    def __init__(self, block):
        super(AttrSet, self).__init__()
        object.__setattr__(self, '_block', None)
        self._block = block
        self._load()

    def _path(self):
        return os.path.join(self._block.basename, 'attr-v2')

    def _load(self):
        path = self._path()
        if not os.path.exists(path):
            return
        with open(path, 'r') as ff:
            for line in ff:
                line = line.split('#', 1)[0].strip()
                if not line:
                    continue
                fields = line.split()
                if len(fields) < 4:
                    continue
                name, dtype_str, nmemb, hexdata = fields[:4]
                try:
                    raw = bytes.fromhex(hexdata)
                    dict.__setitem__(self, name, _decode_attr(
                        dtype_str, int(nmemb), raw))
                except (ValueError, TypeError):
                    continue

    def _flush(self):
        lines = []
        for name in self:
            value = dict.__getitem__(self, name)
            dtype_str, nmemb, raw = _encode_attr(value)
            if isinstance(value, str):
                humane = value
            else:
                humane = ' '.join(repr(v) for v in
                                  numpy.atleast_1d(value).ravel()[:16])
            lines.append('%s %s %d %s #HUMANE [ %s ]\n'
                         % (name, dtype_str, nmemb, raw.hex(), humane))
        with open(self._path(), 'w') as ff:
            ff.writelines(lines)

    def __setitem__(self, name, value):
        # normalize through the codec so reads-after-write are stable
        dtype_str, nmemb, raw = _encode_attr(value)
        dict.__setitem__(self, name, _decode_attr(dtype_str, nmemb, raw))
        self._flush()


# ------------------------------------------------------------------ block
class Block(object):
    """One column of rows (or an attrs-only header block)."""

    def __init__(self, root, name, mode='r'):
        self.name = _normalize(name)
        self.basename = os.path.join(root, *self.name.split('/')) \
            if self.name else root
        self.mode = mode
        if mode == 'r':
            self._read_header()
        self.attrs = AttrSet(self)

    # -- creation ------------------------------------------------------
    @classmethod
    def create(cls, root, name, dtype=None, size=0, Nfile=None):
        self = object.__new__(cls)
        self.name = _normalize(name)
        self.basename = os.path.join(root, *self.name.split('/')) \
            if self.name else root
        self.mode = 'w'
        os.makedirs(self.basename, exist_ok=True)
        if dtype is None:
            self.dtype = None
            self.nmemb = 1
            self.size = 0
            self._file_sizes = []
        else:
            dtype = numpy.dtype(dtype)
            if dtype.shape:
                assert len(dtype.shape) == 1
                self.nmemb = int(dtype.shape[0])
                self.dtype = dtype.base.newbyteorder('<')
            else:
                self.nmemb = 1
                self.dtype = dtype.newbyteorder('<')
            self.size = int(size)
            if Nfile is None:
                Nfile = 1
            if self.size > 0:
                assert Nfile >= 1
            per = self.size // Nfile if Nfile else 0
            self._file_sizes = [per] * Nfile
            for i in range(self.size - per * Nfile):
                self._file_sizes[i] += 1
            # pre-size the physical files so ranks can pwrite regions
            for i, n in enumerate(self._file_sizes):
                path = os.path.join(self.basename, '%06X' % i)
                with open(path, 'wb') as ff:
                    ff.truncate(n * self.itemsize)
        self._write_header()
        self.attrs = AttrSet(self)
        return self

    @property
    def itemsize(self):
        return self.dtype.itemsize * self.nmemb

    def _write_header(self, checksums=None):
        lines = []
        if self.dtype is not None:
            lines.append('DTYPE: %s\n' % self.dtype.str)
            lines.append('NMEMB: %d\n' % self.nmemb)
            lines.append('NFILE: %d\n' % len(self._file_sizes))
            for i, n in enumerate(self._file_sizes):
                ck = 0 if checksums is None else checksums[i]
                lines.append('%06X: %d : %d\n' % (i, n, ck))
        else:
            lines.append('DTYPE: <i8\n')
            lines.append('NMEMB: 1\n')
            lines.append('NFILE: 0\n')
        with open(os.path.join(self.basename, 'header'), 'w') as ff:
            ff.writelines(lines)

    def _read_header(self):
        path = os.path.join(self.basename, 'header')
        if not os.path.exists(path):
            raise KeyError("no bigfile block at %s" % self.basename)
        fields = {}
        files = []
        with open(path, 'r') as ff:
            for line in ff:
                if ':' not in line:
                    continue
                key, _, rest = line.partition(':')
                key = key.strip()
                if key in ('DTYPE', 'NMEMB', 'NFILE'):
                    fields[key] = rest.strip()
                else:
                    files.append((key, int(rest.split(':')[0])))
        self.dtype = numpy.dtype(fields.get('DTYPE', '<i8'))
        self.nmemb = int(fields.get('NMEMB', 1))
        files.sort(key=lambda kv: int(kv[0], 16))
        self._file_sizes = [n for _, n in files]
        self.size = sum(self._file_sizes)

    # -- data ----------------------------------------------------------
    def _file_start(self, i):
        return sum(self._file_sizes[:i])

    def write(self, offset, array):
        """Write ``len(array)`` rows at row ``offset`` (casting to the
        block dtype, as the reference relies on: io tests write f8 data
        into f4 blocks)."""
        assert self.dtype is not None, "attrs-only block"
        shape = (len(array), self.nmemb) if self.nmemb > 1 \
            else (len(array),)
        array = numpy.ascontiguousarray(
            numpy.asarray(array).reshape(shape), dtype=self.dtype)
        raw = array.tobytes()
        isz = self.itemsize
        row = int(offset)
        done = 0
        nrows = len(array)
        for i, n in enumerate(self._file_sizes):
            fstart = self._file_start(i)
            lo = max(row, fstart)
            hi = min(row + nrows, fstart + n)
            if lo >= hi:
                continue
            path = os.path.join(self.basename, '%06X' % i)
            with open(path, 'r+b') as ff:
                ff.seek((lo - fstart) * isz)
                ff.write(raw[(lo - row) * isz:(hi - row) * isz])
            done += hi - lo
        if done != nrows:
            raise IndexError("write of %d rows at %d exceeds block size "
                             "%d" % (nrows, row, self.size))

    def update_checksums(self):
        """Recompute the per-file sysv sums into the header.  Not called
        on the write path: no reader (the reference's included) verifies
        the field, and recomputing it would re-read every byte written —
        callers wanting the sums filled in invoke this explicitly."""
        checksums = []
        for i, n in enumerate(self._file_sizes):
            path = os.path.join(self.basename, '%06X' % i)
            with open(path, 'rb') as ff:
                checksums.append(_sysv_checksum(ff.read()))
        self._write_header(checksums)

    def __getitem__(self, sl):
        assert self.dtype is not None, "attrs-only block"
        if isinstance(sl, slice):
            start, stop, step = sl.indices(self.size)
        else:
            raise TypeError("block reads use slices")
        out = numpy.empty((max(0, stop - start), self.nmemb),
                          dtype=self.dtype)
        isz = self.itemsize
        for i, n in enumerate(self._file_sizes):
            fstart = self._file_start(i)
            lo = max(start, fstart)
            hi = min(stop, fstart + n)
            if lo >= hi:
                continue
            path = os.path.join(self.basename, '%06X' % i)
            with open(path, 'rb') as ff:
                ff.seek((lo - fstart) * isz)
                raw = ff.read((hi - lo) * isz)
            out[lo - start:hi - start] = numpy.frombuffer(
                raw, dtype=self.dtype).reshape(hi - lo, self.nmemb)
        out = out[::step]
        if self.nmemb == 1:
            return out[:, 0]
        return out

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        return False


# ------------------------------------------------------------------ file
class BigFile(object):
    """The directory container (the subset of ``bigfile.File`` /
    ``bigfile.FileMPI`` the reference call sites use).  ``comm`` makes
    ``create``/``create_from_array`` collective: rank 0 creates, every
    rank writes its own row range."""

    def __init__(self, filename, create=False, comm=None):
        self.filename = filename
        self.comm = comm
        if create and self._rank() == 0:
            os.makedirs(filename, exist_ok=True)
        self._barrier()
        if not os.path.isdir(filename):
            raise IOError("no bigfile directory at %r" % filename)

    def _rank(self):
        return 0 if self.comm is None else self.comm.rank

    def _barrier(self):
        if self.comm is not None:
            self.comm.barrier()

    @property
    def blocks(self):
        found = []
        root = os.path.abspath(self.filename)
        for dirpath, dirnames, filenames in os.walk(root):
            if 'header' in filenames:
                rel = os.path.relpath(dirpath, root)
                found.append('.' if rel == '.' else
                             rel.replace(os.sep, '/'))
            dirnames.sort()
        return sorted(found)

    def __getitem__(self, name):
        if name.endswith('/'):
            sub = BigFile.__new__(BigFile)
            sub.filename = os.path.join(self.filename,
                                        *_normalize(name).split('/'))
            sub.comm = self.comm
            return sub
        return Block(self.filename, name)

    def __contains__(self, name):
        try:
            Block(self.filename, name)
            return True
        except KeyError:
            return False

    def create(self, name, dtype=None, size=0, Nfile=1):
        if self._rank() == 0:
            block = Block.create(self.filename, name, dtype=dtype,
                                 size=size, Nfile=Nfile)
        self._barrier()
        if self._rank() != 0:
            block = Block(self.filename, name)
        return block

    def create_from_array(self, name, array, comm_offsets=None):
        """Create sized for the global array and write this rank's rows
        (``MeshSource.save``, reference base/mesh.py:444-480).  With a
        comm, ``array`` is the local part; rows are ordered by rank."""
        array = numpy.asarray(array)
        nlocal = len(array)
        if self.comm is None:
            size, offset = nlocal, 0
        else:
            sizes = self.comm.allgather(nlocal)
            size = sum(sizes)
            offset = sum(sizes[:self.comm.rank])
        dtype = numpy.dtype((array.dtype, array.shape[1:]))
        sizeperfile = 32 * 1024 * 1024
        Nfile = max(1, (size + sizeperfile - 1) // sizeperfile)
        block = self.create(name, dtype=dtype, size=size, Nfile=Nfile)
        # ranks write disjoint regions; serialize for the shared header
        for r in range(1 if self.comm is None else self.comm.size):
            if r == self._rank() and nlocal > 0:
                block.write(offset, array)
            self._barrier()
        # fill the per-file sysv sums so headers interchange with tools
        # that diff/validate them (bigfile's C writer always records them)
        if self._rank() == 0:
            block.update_checksums()
        self._barrier()
        return block

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        return False


FileMPI = BigFile   # the reference imports FileMPI for collective io


class Dataset(object):
    """Column-concatenated view over sibling blocks (``bigfile.Dataset``
    as used by io/bigfile.py:77-96): structured ``dtype``, common
    ``size``, per-column and structured slice reads."""

    def __init__(self, subfile, columns):
        self.file = subfile
        self.columns = sorted(columns)
        self._blocks = {c: subfile[c] for c in self.columns}
        sizes = set(b.size for b in self._blocks.values())
        if len(sizes) > 1:
            raise ValueError("dataset columns have unequal sizes: %s"
                             % {c: b.size for c, b in
                                self._blocks.items()})
        self.size = sizes.pop() if sizes else 0
        specs = []
        for c in self.columns:
            b = self._blocks[c]
            specs.append((c, (b.dtype, (b.nmemb,)) if b.nmemb > 1
                          else b.dtype))
        self.dtype = numpy.dtype(specs)

    def __getitem__(self, key):
        if isinstance(key, str):
            return self._blocks[key]
        out = numpy.empty(len(range(*key.indices(self.size))),
                          dtype=self.dtype)
        for c in self.columns:
            out[c] = self._blocks[c][key]
        return out
