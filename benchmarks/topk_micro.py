import time

import torch


def bench(fn, name, iters=10):
    for _ in range(2):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    print(f"{name:44s} {(time.perf_counter()-t0)/iters*1000:8.2f} ms")


n, nq = 5_000_000, 64
s = torch.randn(n, nq, device="cuda")
bench(lambda: torch.topk(s, 10, dim=0), "topk k=10 dim=0 of (n, nq)")
bench(lambda: torch.topk(s.T.contiguous(), 10, dim=1), "transpose+contig then topk dim=1")
st = s.T.contiguous()
bench(lambda: torch.topk(st, 10, dim=1), "topk k=10 dim=1 of (nq, n) contig")
bench(lambda: s.T.contiguous(), "transpose+contig alone")
bench(lambda: torch.topk(st, 160, dim=1), "topk k=160 dim=1 contig")
