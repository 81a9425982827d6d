#!/usr/bin/env python3
"""Isolated ANN kernel probe for rocprofv3 counter runs."""
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from lakesoul_amd.ops import hip

n, dim, nq = 1_000_000, 768, 64
rng = np.random.default_rng(0)
X = torch.from_numpy(rng.normal(size=(n, dim)).astype(np.float32)).to(torch.bfloat16).cuda()
Q = torch.from_numpy(rng.normal(size=(nq, dim)).astype(np.float32)).to(torch.bfloat16).cuda()
hip().ann_scores(X, Q)
torch.cuda.synchronize()
t0 = time.time()
iters = 20
for _ in range(iters):
    s = hip().ann_scores(X, Q)
torch.cuda.synchronize()
dt = (time.time() - t0) / iters
flops = 2.0 * n * nq * dim
print(f"ann_scores kernel: {dt*1000:.2f} ms/call, {flops/dt/1e12:.1f} TFLOP/s, "
      f"X-read {(n*dim*2)*(nq//64 if nq>=64 else 1)/dt/1e9:.0f} GB/s")
