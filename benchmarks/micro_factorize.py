import torch, time

def bench(fn, name, iters=20):
    for _ in range(3): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    print(f"{name:40s} {(time.perf_counter()-t0)/iters*1000:8.3f} ms")

n = 6_000_000
a8 = torch.randint(0, 3, (n,), dtype=torch.int8, device="cuda")
b8 = torch.randint(0, 2, (n,), dtype=torch.int8, device="cuda")
a64 = a8.to(torch.int64); b64 = b8.to(torch.int64)

bench(lambda: torch.unique(a8, sorted=True, return_inverse=True), "unique int8")
bench(lambda: torch.unique(a64, sorted=True, return_inverse=True), "unique int64")
bench(lambda: torch.unique(a64*2+b64, sorted=True, return_inverse=True), "unique combined int64")
bench(lambda: torch.sort(a64), "sort int64")
bench(lambda: torch.bincount(a64, minlength=3), "bincount 3 slots")
bench(lambda: torch.bincount(a64*2+b64, minlength=65536), "bincount 64k slots")
def dense_lut():
    comb = a64*2+b64
    cnt = torch.bincount(comb, minlength=6)
    used = torch.nonzero(cnt, as_tuple=True)[0]
    lut = torch.zeros(6, dtype=torch.int64, device="cuda")
    lut[used] = torch.arange(used.numel(), device="cuda")
    return lut[comb]
bench(dense_lut, "dense bincount+lut (6 slots)")
def masks_used():
    comb = a64*2+b64
    m = comb == torch.arange(6, device="cuda")[:,None]
    cnts = m.sum(1)
    used = torch.nonzero(cnts, as_tuple=True)[0]
    lut = torch.zeros(6, dtype=torch.int64, device="cuda")
    lut[used] = torch.arange(used.numel(), device="cuda")
    return lut[comb]
bench(masks_used, "dense masks+lut (6 slots)")
def amin_amax():
    return int(a64.min()), int(a64.max())
bench(amin_amax, "min+max int64 (sync)")

codes = (a64*2+b64)
def rep_scatter():
    rep = torch.full((6,), n, dtype=torch.int64, device="cuda")
    rep.scatter_reduce_(0, codes, torch.arange(n, dtype=torch.int64, device="cuda"),
                        reduce="amin", include_self=True)
    return rep
bench(rep_scatter, "rep via scatter_reduce amin (6 slots)")
def rep_sort():
    order = torch.argsort(codes, stable=True)
    cs = codes[order]
    first = torch.ones_like(cs, dtype=torch.bool)
    first[1:] = cs[1:] != cs[:-1]
    return order[first]
bench(rep_sort, "rep via stable argsort boundaries")
