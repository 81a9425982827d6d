#!/usr/bin/env python3
"""Vector ANN recall@10 / QPS table: exact MFMA vs 1-bit binary vs
IVF-RaBitQ fastscan (VERDICT r1 #4 done-criterion: the RaBitQ pipeline
must beat the 1-bit pass at >=5M x 768-d).

Run (GPU box):
  python benchmarks/vector_bench.py --n 5000000 --dim 768 --queries 64
Prints one JSON line per engine: {engine, recall_at_10, qps, build_s}.
"""

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=5_000_000)
    p.add_argument("--dim", type=int, default=768)
    p.add_argument("--queries", type=int, default=64)
    p.add_argument("--k", type=int, default=10)
    p.add_argument("--clusters", type=int, default=0)
    p.add_argument("--nprobe", type=int, default=0)
    p.add_argument("--repeat", type=int, default=3)
    p.add_argument("--device", default="cuda" if torch.cuda.is_available() else "cpu")
    args = p.parse_args()
    dev = args.device
    n, dim, k = args.n, args.dim, args.k
    clusters = args.clusters or int(min(4096, max(64, n // 4096)))
    nprobe = args.nprobe or max(8, clusters // 16)

    rng = np.random.default_rng(0)
    # clustered synthetic embeddings (gaussian mixture, normalized)
    n_cent = 1024
    cents = rng.normal(size=(n_cent, dim)).astype(np.float32)
    print(f"[vb] generating {n}x{dim}...", file=sys.stderr, flush=True)
    # generate on GPU in chunks to bound host RAM
    vecs_t = torch.empty(n, dim, dtype=torch.float32, device=dev)
    g = torch.Generator(device="cpu").manual_seed(1)
    cents_t = torch.from_numpy(cents).to(dev)
    chunk = 1_000_000
    for a in range(0, n, chunk):
        b = min(a + chunk, n)
        assign = torch.randint(0, n_cent, (b - a,), generator=g)
        noise = torch.randn(b - a, dim, generator=g)
        v = cents_t[assign.to(dev)] + 0.6 * noise.to(dev)
        vecs_t[a:b] = v / v.norm(dim=1, keepdim=True)
    qsel = torch.randint(0, n, (args.queries,), generator=g)
    qs = (vecs_t[qsel.to(dev)] +
          0.3 * torch.randn(args.queries, dim, generator=g).to(dev))
    qs = qs / qs.norm(dim=1, keepdim=True)

    from lakesoul_amd.vector.rabitq import (
        compute_const_scaling_factor, quantize_batch)
    from lakesoul_amd.vector.index import kmeans

    # ---- ground truth: exact bf16 on MFMA ---- #
    from lakesoul_amd.ops import hip as hip_mod
    vb = vecs_t.to(torch.bfloat16)

    def exact_scores(q):
        if str(dev).startswith("cuda"):
            nq = q.shape[0]
            nq_pad = (nq + 15) // 16 * 16
            qb = torch.zeros(nq_pad, dim, dtype=torch.bfloat16, device=dev)
            qb[:nq] = q.to(torch.bfloat16)
            return hip_mod().ann_scores_t(vb, qb)[:nq]   # (nq, n)
        return (vecs_t @ q.T).T.contiguous()

    torch.cuda.synchronize() if dev == "cuda" else None
    t0 = time.time()
    sc = exact_scores(qs)
    truth = torch.topk(sc, k, dim=1).indices    # (nq, k)
    torch.cuda.synchronize() if dev == "cuda" else None
    exact_s = time.time() - t0
    truth_sets = [set(truth[i].tolist()) for i in range(args.queries)]

    # timed exact QPS
    reps = args.repeat
    t0 = time.time()
    for _ in range(reps):
        s = exact_scores(qs)
        torch.topk(s, k, dim=1)
    torch.cuda.synchronize() if dev == "cuda" else None
    qps_exact = args.queries * reps / (time.time() - t0)
    print(json.dumps({"engine": "exact-mfma-bf16", "recall_at_10": 1.0,
                      "qps": qps_exact, "n": n, "dim": dim,
                      "first_query_s": exact_s}), flush=True)

    # ---- IVF clustering (shared) ---- #
    t0 = time.time()
    sample = vecs_t[torch.randperm(n, generator=g)[:200_000].to(dev)]
    centroids = kmeans(sample, clusters, iters=8)
    assign = torch.empty(n, dtype=torch.int64, device=dev)
    for a in range(0, n, chunk):
        b = min(a + chunk, n)
        assign[a:b] = (vecs_t[a:b] @ centroids.T).argmax(dim=1)
    order = torch.argsort(assign, stable=True)
    vecs_s = vecs_t[order]
    vb_s = vecs_s.to(torch.bfloat16)
    assign_s = assign[order]
    counts = torch.bincount(assign_s, minlength=clusters)
    clu = torch.zeros(clusters + 1, dtype=torch.int64)
    torch.cumsum(counts.cpu(), 0, out=clu[1:].view(-1))
    ivf_s = time.time() - t0

    # ---- RaBitQ quantization (4 bits: 1 sign + 3 ex) ---- #
    t0 = time.time()
    eb = 3
    t_const = compute_const_scaling_factor(dim, eb)
    bits_l, ex_l, fac_l = [], [], []
    for cid in range(clusters):
        a, b = int(clu[cid]), int(clu[cid + 1])
        if b <= a:
            continue
        qb = quantize_batch(vecs_s[a:b], centroids[cid], eb, t_const, "ip")
        bits_l.append(qb.bits_packed)
        ex_l.append(qb.ex_packed)
        fac_l.append(torch.stack([qb.f_add, qb.f_rescale, qb.f_error,
                                  qb.f_add_ex, qb.f_rescale_ex, qb.delta, qb.vl]))
    bits_all = torch.cat(bits_l).contiguous()
    ex_all = torch.cat(ex_l).contiguous()
    fac_all = torch.cat(fac_l, dim=1).contiguous()
    quant_s = time.time() - t0

    sum_q = qs.sum(dim=1)
    c1_sum_q = -0.5 * sum_q
    cb = -((1 << eb) - 0.5)
    cb_sum_q = cb * sum_q
    bscale = float(1 << eb)
    g_add_all = -(qs @ centroids.T)  # ip metric

    cl_of_row = torch.repeat_interleave(
        torch.arange(clusters, dtype=torch.int64), counts.cpu()).to(dev)

    def staged_search(use_ex: bool, C: int, R: int, probe_frac: float = 1.0):
        """Single-pass fastscan over ALL rows (96 B/vec HBM traffic), est
        masked to the probed clusters when probe_frac < 1."""
        from lakesoul_amd.utils import timing as _tm

        ip_T = None
        if str(dev).startswith("cuda"):
            with _tm.phase("vb_fastscan_est", sync_gpu=True):
                est = hip_mod().fastscan_est(
                    bits_all, qs, dim, fac_all[0].contiguous(),
                    fac_all[1].contiguous(), cl_of_row.to(torch.int32),
                    g_add_all.contiguous(), c1_sum_q.contiguous())  # (nq, n)
        else:
            from lakesoul_amd.vector.rabitq import unpack_bits

            ip = unpack_bits(bits_all, dim).to(torch.float32) @ qs.T
            ip_T = ip.T
            est = (fac_all[0][None, :] + g_add_all[:, cl_of_row]
                   + fac_all[1][None, :] * (ip_T + c1_sum_q[:, None]))
        if probe_frac < 1.0:
            npb = max(1, int(clusters * probe_frac))
            cprobe = torch.topk(qs @ centroids.T, npb, dim=1).indices
            probe_mask = torch.zeros(args.queries, clusters, dtype=torch.bool,
                                     device=dev)
            probe_mask.scatter_(1, cprobe, True)
            est = torch.where(probe_mask[:, cl_of_row], est,
                              torch.full_like(est, float("inf")))
        with _tm.phase("vb_stage1_topk", sync_gpu=True):
            top_c = torch.topk(-est, min(C, n), dim=1)
        cand_est = -top_c.values
        cand_row = top_c.indices
        if ip_T is not None:
            cand_ip = torch.gather(ip_T, 1, cand_row)
        else:
            fr = fac_all[1][cand_row]
            ga = torch.gather(g_add_all, 1, cl_of_row[cand_row])
            cand_ip = torch.where(
                fr != 0,
                (cand_est - fac_all[0][cand_row] - ga) / fr - c1_sum_q[:, None],
                torch.zeros_like(cand_est))
        _ex = _tm.phase("vb_ex_refine", sync_gpu=True)
        _ex.__enter__()
        if use_ex:
            rows = cand_row
            if str(dev).startswith("cuda"):
                # pair-wise: one dot per (query, candidate) — no unique /
                # gather round trip, no nq-x redundant dots
                ex_dot = hip_mod().fastscan_ex_dot_pairs(ex_all, rows, qs, dim)
            else:
                from lakesoul_amd.vector.rabitq import unpack_nibbles

                flat = torch.unique(rows.flatten())
                exd = unpack_nibbles(ex_all[flat], dim).to(torch.float32) @ qs.T
                pos = torch.searchsorted(flat, rows.flatten()).view(rows.shape)
                ex_dot = torch.gather(exd.T, 1, pos)
            g_add_cand = torch.gather(g_add_all, 1, cl_of_row[rows])
            tt = bscale * cand_ip + ex_dot + cb_sum_q[:, None]
            refined = fac_all[3, rows] + g_add_cand + fac_all[4, rows] * tt
            refined = torch.where(cand_row >= 0, refined,
                                  torch.full_like(refined, float("inf")))
        else:
            refined = cand_est
        _ex.__exit__(None, None, None)
        rr = min(R, refined.shape[1])
        _rs = _tm.phase("vb_rescore", sync_gpu=True)
        _rs.__enter__()
        topr = torch.topk(-refined, rr, dim=1).indices
        rrows = torch.gather(cand_row, 1, topr).clamp_min(0)
        # exact rescore of R rows per query
        flat = torch.unique(rrows.flatten())
        sub = vb_s[flat].to(torch.float32) @ qs.T
        pos = torch.searchsorted(flat, rrows.flatten()).view(rrows.shape)
        exact = torch.gather(sub.T, 1, pos)
        top = torch.topk(exact, k, dim=1).indices
        final_rows = torch.gather(rrows, 1, top)
        _rs.__exit__(None, None, None)
        return order[final_rows.flatten()].view(final_rows.shape)

    cases = [
        ("fastscan-1bit+rescore", False, 256 * k, 32 * k, 1.0),
        ("fastscan-rabitq4+rescore", True, 256 * k, 16 * k, 1.0),
        ("fastscan-rabitq4-hiC", True, 1024 * k, 16 * k, 1.0),
        ("ivf25-rabitq4+rescore", True, 256 * k, 16 * k, 0.25),
    ]
    for name, use_ex, C, R, pf in cases:
        ids = staged_search(use_ex, C, R, pf)
        recall = np.mean([
            len(truth_sets[i] & set(ids[i].tolist())) / k
            for i in range(args.queries)])
        torch.cuda.synchronize() if dev == "cuda" else None
        t0 = time.time()
        for _ in range(reps):
            staged_search(use_ex, C, R, pf)
        torch.cuda.synchronize() if dev == "cuda" else None
        qps = args.queries * reps / (time.time() - t0)
        print(json.dumps({"engine": name, "recall_at_10": float(recall),
                          "qps": qps, "n": n, "dim": dim,
                          "clusters": clusters, "probe_frac": pf,
                          "C": C, "R": R,
                          "ivf_s": ivf_s, "quant_s": quant_s}), flush=True)
        from lakesoul_amd.utils import timing as _tmr

        if _tmr.ENABLED:
            print(f"-- stage timing ({name}):\n{_tmr.report()}", flush=True)
            _tmr._acc.clear()
            _tmr._cnt.clear()


if __name__ == "__main__":
    main()
