#!/usr/bin/env python3
"""ANN vector search benchmark (BASELINE config 5): 768-d embeddings,
MFMA bf16 cosine scoring + top-k over per-bucket shards.

    python benchmarks/ann_bench.py --n 1000000 --dim 768 --nq 64 --k 10
"""

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=1_000_000)
    p.add_argument("--dim", type=int, default=768)
    p.add_argument("--nq", type=int, default=64)
    p.add_argument("--k", type=int, default=10)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--device", default=None)
    p.add_argument("--binary", action="store_true",
                   help="1-bit sign-code first pass + MFMA rescore")
    p.add_argument("--rescore", type=int, default=16)
    args = p.parse_args()
    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")

    rng = np.random.default_rng(0)
    X = torch.from_numpy(rng.normal(size=(args.n, args.dim)).astype(np.float32))
    X = X / X.norm(dim=1, keepdim=True)
    Xb = X.to(torch.bfloat16).to(device)
    Q = torch.from_numpy(rng.normal(size=(args.nq, args.dim)).astype(np.float32))
    Q = (Q / Q.norm(dim=1, keepdim=True)).to(torch.bfloat16).to(device)

    codes = qcodes = None
    if args.binary:
        from lakesoul_amd.vector.index import pack_sign_bits, random_rotation

        rot = random_rotation(args.dim)
        codes = torch.from_numpy(pack_sign_bits(X.numpy() @ rot)).to(device)
        qn = Q.to(torch.float32).cpu().numpy() @ rot
        qcodes = torch.from_numpy(pack_sign_bits(qn)).to(device)

    def search():
        from lakesoul_amd.ops import hip

        if args.binary and device == "cuda":
            ham = hip().hamming_scores(codes, qcodes)
            nc = min(args.k * args.rescore, args.n)
            cand = torch.topk(-ham.to(torch.float16), nc, dim=0).indices
            flat = torch.unique(cand.flatten())
            sub = hip().ann_scores(Xb[flat].contiguous(), Q)
            top = torch.topk(sub, args.k, dim=0)
            return flat[top.indices]
        if device == "cuda":
            scores = hip().ann_scores(Xb, Q)
        else:
            scores = Xb.to(torch.float32) @ Q.to(torch.float32).T
        return torch.topk(scores, args.k, dim=0)

    # self-recall sanity
    qi = rng.choice(args.n, args.nq, replace=False)
    Qs = Xb[qi]
    if device == "cuda":
        from lakesoul_amd.ops import hip

        s = hip().ann_scores(Xb, Qs)
    else:
        s = Xb.to(torch.float32) @ Qs.to(torch.float32).T
    top1 = s.argmax(dim=0).cpu().numpy()
    recall = float(np.mean(top1 == qi))

    search()  # warmup
    if device == "cuda":
        torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(args.steps):
        out = search()
    if device == "cuda":
        torch.cuda.synchronize()
    dt = (time.time() - t0) / args.steps

    flops = 2.0 * args.n * args.nq * args.dim
    gbytes = args.n * args.dim * 2 / 1e9  # X read once per query tile group
    print(
        json.dumps(
            {
                "metric": "ann_cosine_topk",
                "n": args.n,
                "dim": args.dim,
                "nq": args.nq,
                "k": args.k,
                "s_per_search": dt,
                "vectors_per_sec": args.n / dt,
                "tflops": flops / dt / 1e12,
                "x_read_gb_per_s": gbytes * ((args.nq + 15) // 16) / dt,
                "self_recall_at_1": recall,
                "binary_first_pass": bool(args.binary),
                "device": device,
            }
        )
    )


if __name__ == "__main__":
    main()
