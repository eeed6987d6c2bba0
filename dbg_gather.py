import numpy, torch
from nbodykit_amd import set_options, hiplib
from nbodykit_amd.source.mesh.catalog import _prepare_particles
from nbodykit_amd.pm import ParticleMesh

rng = numpy.random.RandomState(17)
n = 200000
pos = numpy.concatenate([rng.uniform(0, 64., size=(n, 3)),
                         rng.normal(32., 1.5, size=(n // 4, 3)) % 64.])
rng.shuffle(pos)
pm = ParticleMesh(BoxSize=64., Nmesh=[64, 64, 64], dtype='f8')
pos_t = torch.as_tensor(pos).to('cuda')

with set_options(sort_min_n=1024, sort_two_level_min_n=1024,
                 sort_two_level_min_cells=1):
    soa, mass, sorted_, rowtab = _prepare_particles(pos_t, None, pm)
print("rowtab is None?", rowtab is None, "sorted:", sorted_)

N = 64
cells = ((numpy.floor(pos * (N / 64.)).astype('i8')) % N)
cellid = (cells[:, 0] * N + cells[:, 1]) * N + cells[:, 2]

x = soa[:len(pos)].cpu().numpy()
y = soa[len(pos):2*len(pos)].cpu().numpy()
z = soa[2*len(pos):].cpu().numpy()
c2 = ((numpy.floor(numpy.stack([x, y, z], 1) * (N / 64.)).astype('i8')) % N)
cid2 = (c2[:, 0] * N + c2[:, 1]) * N + c2[:, 2]
print("sorted output cell-ordered:", bool((numpy.diff(cid2) >= 0).all()))
print("same multiset:", numpy.array_equal(numpy.sort(cellid), cid2) )

# rowtab check
rt = rowtab.cpu().numpy()
rowid = cells[:, 0] * N + cells[:, 1]
expect = numpy.searchsorted(numpy.sort(rowid), numpy.arange(N * N))
print("rowtab ok:", numpy.array_equal(rt[:-1], expect), "sentinel:", rt[-1] == len(pos))
if not numpy.array_equal(rt[:-1], expect):
    bad = numpy.nonzero(rt[:-1] != expect)[0]
    print("first bad rows:", bad[:5], rt[bad[:5]], expect[bad[:5]])

# direct vs gather paint
lib = hiplib.require()
nmesh = hiplib.i64_arr(pm.Nmesh); box = hiplib.f64_arr(pm.BoxSize)
mesh1 = torch.zeros((N, N, N), dtype=torch.float64, device='cuda')
hiplib.check(lib.nbk_paint_f64(hiplib.dptr(soa), None, len(pos), nmesh, box,
    0, 0.0, hiplib.dptr(mesh1), 0, N, hiplib.cur_stream()), 'p')
mesh2 = torch.zeros((N, N, N), dtype=torch.float64, device='cuda')
hiplib.check(lib.nbk_paint_gather_f64(hiplib.dptr(soa), None, len(pos), nmesh,
    box, 0, 0.0, hiplib.dptr(rowtab), hiplib.dptr(mesh2), 0, N, 0,
    hiplib.cur_stream()), 'g')
torch.cuda.synchronize()
d = (mesh2 - mesh1).abs()
print("max abs diff:", d.max().item(), "sum1:", mesh1.sum().item(), "sum2:", mesh2.sum().item())
idx = d.argmax().item()
print("argmax cell:", numpy.unravel_index(idx, (N,N,N)), "v1:",
      mesh1.view(-1)[idx].item(), "v2:", mesh2.view(-1)[idx].item())
