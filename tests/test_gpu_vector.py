"""GPU MFMA ANN tests: numerics vs plain fp32 torch reference."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


def test_ann_scores_mfma_vs_fp32(dev):
    """mfma_f32_16x16x32_bf16 scoring against a plain PyTorch fp32
    reference — random asymmetric inputs (guide G9)."""
    from lakesoul_amd.ops import hip

    rng = np.random.default_rng(0)
    for n, nq, K in [(1024, 16, 768), (777, 32, 128), (4096, 64, 768)]:
        X = torch.from_numpy(rng.normal(size=(n, K)).astype(np.float32))
        Q = torch.from_numpy(rng.normal(size=(nq, K)).astype(np.float32))
        Xb = X.to(torch.bfloat16).to(dev)
        Qb = Q.to(torch.bfloat16).to(dev)
        got = hip().ann_scores(Xb, Qb).cpu()  # (n, nq)
        ref = (Xb.to(torch.float32) @ Qb.to(torch.float32).T).cpu()
        torch.testing.assert_close(got, ref, rtol=1e-3, atol=1e-2)


def test_end_to_end_vector_search_gpu(dev, tmp_path):
    from lakesoul_amd.meta.client import MetaClient
    from lakesoul_amd.meta.store import SqliteMetaStore
    from lakesoul_amd.tables.catalog import LakeSoulCatalog
    from lakesoul_amd.io.schema import Field, Schema
    from lakesoul_amd.vector.index import build_vector_index

    catalog = LakeSoulCatalog(
        MetaClient(SqliteMetaStore(str(tmp_path / "meta.db"))),
        warehouse=str(tmp_path / "wh"),
    )
    rng = np.random.default_rng(1)
    n, dim = 50000, 768
    t = catalog.create_table(
        "gvec",
        Schema([Field("id", "int64", False), Field("emb", "binary", False)]),
        primary_keys=["id"],
        hash_bucket_num=4,
    )
    vecs = rng.normal(size=(n, dim)).astype(np.float32)
    t.upsert({"id": np.arange(n, dtype=np.int64), "emb": [v.tobytes() for v in vecs]})
    idx = build_vector_index(t, "emb", metric="cosine")
    qids = rng.choice(n, 32, replace=False)
    ids, scores = idx.search(vecs[qids], k=10, device="cuda")
    recall1 = float(np.mean(ids[:, 0] == qids))
    assert recall1 >= 0.95, f"self-recall@1 {recall1}"
    # cross-check against the CPU search
    ids_cpu, _ = idx.search(vecs[qids[:8]], k=10, device="cpu")
    overlap = np.mean([
        len(set(ids[i, :10]) & set(ids_cpu[i, :10])) / 10.0 for i in range(8)
    ])
    assert overlap >= 0.85


@pytest.mark.gpu
def test_gpu_hamming_kernel_matches_numpy(dev):
    """hamming_scores kernel vs numpy bitwise_count."""
    from lakesoul_amd.ops import hip

    rng = np.random.default_rng(0)
    n, nq, w = 5000, 33, 12
    codes = rng.integers(-(2**62), 2**62, (n, w)).astype(np.int64)
    qc = rng.integers(-(2**62), 2**62, (nq, w)).astype(np.int64)
    out = hip().hamming_scores(torch.from_numpy(codes).to(dev),
                               torch.from_numpy(qc).to(dev)).cpu().numpy()
    ref = np.bitwise_count(
        codes.view(np.uint64)[:, None, :] ^ qc.view(np.uint64)[None, :, :]
    ).sum(axis=2).astype(np.int32)
    np.testing.assert_array_equal(out, ref)


@pytest.mark.gpu
def test_gpu_binary_index_recall(dev, tmp_path):
    """Binary first pass + MFMA rescore on GPU matches exact top-1."""
    from lakesoul_amd.meta.client import MetaClient
    from lakesoul_amd.meta.store import SqliteMetaStore
    from lakesoul_amd.tables.catalog import LakeSoulCatalog
    from lakesoul_amd.io.schema import Field, Schema
    from lakesoul_amd.vector.index import build_vector_index

    cat = LakeSoulCatalog(MetaClient(SqliteMetaStore(str(tmp_path / "m.db"))),
                          warehouse=str(tmp_path / "wh"))
    rng = np.random.default_rng(5)
    n, dim = 20000, 128
    t = cat.create_table(
        "gbin", Schema([Field("id", "int64", False), Field("emb", "binary", False)]),
        primary_keys=["id"], hash_bucket_num=2)
    vecs = rng.normal(size=(n, dim)).astype(np.float32)
    t.upsert({"id": np.arange(n, dtype=np.int64), "emb": [v.tobytes() for v in vecs]})
    exact = build_vector_index(t, "emb", metric="cosine")
    qids = rng.choice(n, 32, replace=False)
    ids_e, _ = exact.search(vecs[qids], k=10, device="cuda")
    idx = build_vector_index(t, "emb", metric="cosine", binary=True)
    # gaussian-random vectors are the hardest case for 1-bit codes (no
    # cluster structure, weakly separated neighbors): go deeper on rescore
    ids_b, _ = idx.search(vecs[qids], k=10, device="cuda", rescore=64)
    recall = np.mean([len(set(ids_b[i]) & set(ids_e[i])) / 10.0 for i in range(32)])
    assert recall >= 0.9, recall
    assert (ids_b[:, 0] == qids).mean() >= 0.95


def test_fastscan_bit_dot_vs_oracle(dev):
    """LDS-LUT 1-bit dot kernel vs the torch unpack oracle
    (rabitq.py estimate path)."""
    from lakesoul_amd.ops import hip
    from lakesoul_amd.vector.rabitq import pack_bits, unpack_bits

    rng = np.random.default_rng(7)
    for m, nq, dim in [(1000, 4, 64), (4097, 7, 768), (333, 1, 100)]:
        bits = torch.from_numpy(rng.integers(0, 2, (m, dim)).astype(np.uint8))
        packed = pack_bits(bits.bool()).to(dev)
        q = torch.from_numpy(rng.normal(size=(nq, dim)).astype(np.float32)).to(dev)
        got = hip().fastscan_bit_dot(packed, q, dim).cpu()
        ref = (bits.to(torch.float32) @ q.cpu().T)
        torch.testing.assert_close(got, ref, rtol=1e-4, atol=1e-3)


def test_fastscan_est_fused_vs_oracle(dev):
    """Fused estimator kernel vs the torch composition of the same
    quantities."""
    from lakesoul_amd.ops import hip
    from lakesoul_amd.vector.rabitq import pack_bits, unpack_bits

    rng = np.random.default_rng(9)
    m, nq, dim, kc = 3000, 5, 256, 16
    bits = torch.from_numpy(rng.integers(0, 2, (m, dim)).astype(np.uint8))
    packed = pack_bits(bits.bool()).to(dev)
    q = torch.from_numpy(rng.normal(size=(nq, dim)).astype(np.float32)).to(dev)
    f_add = torch.from_numpy(rng.normal(size=m).astype(np.float32)).to(dev)
    f_res = torch.from_numpy(rng.normal(size=m).astype(np.float32)).to(dev)
    cl = torch.from_numpy(rng.integers(0, kc, m).astype(np.int32)).to(dev)
    g_add = torch.from_numpy(rng.normal(size=(nq, kc)).astype(np.float32)).to(dev)
    c1sq = (-0.5 * q.sum(dim=1)).contiguous()
    got = hip().fastscan_est(packed, q, dim, f_add, f_res, cl, g_add, c1sq).cpu()
    ip = (bits.to(torch.float32) @ q.cpu().T).T  # (nq, m)
    ref = (f_add.cpu()[None, :] + torch.gather(
        g_add.cpu(), 1, cl.cpu().to(torch.int64)[None, :].expand(nq, m))
        + f_res.cpu()[None, :] * (ip + c1sq.cpu()[:, None]))
    # f16 LDS LUTs in the query-blocked kernel: ~5e-4 relative per LUT
    # entry — far under the estimate's own 1-bit quantization error
    torch.testing.assert_close(got, ref, rtol=5e-3, atol=5e-2)


def test_fastscan_ex_dot_vs_oracle(dev):
    from lakesoul_amd.ops import hip
    from lakesoul_amd.vector.rabitq import pack_nibbles, unpack_nibbles

    rng = np.random.default_rng(8)
    for m, nq, dim in [(500, 3, 64), (2049, 5, 768), (100, 2, 99)]:
        codes = torch.from_numpy(rng.integers(0, 8, (m, dim)).astype(np.uint8))
        packed = pack_nibbles(codes).to(dev)
        q = torch.from_numpy(rng.normal(size=(nq, dim)).astype(np.float32)).to(dev)
        got = hip().fastscan_ex_dot(packed, q, dim).cpu()
        ref = codes.to(torch.float32) @ q.cpu().T
        torch.testing.assert_close(got, ref, rtol=1e-4, atol=1e-3)


def test_fastscan_ex_dot_pairs_vs_oracle(dev):
    """Pair-wise ex dots (one dot per (query, candidate)) match the
    dense oracle on the selected pairs; -1 rows produce 0."""
    from lakesoul_amd.ops import hip
    from lakesoul_amd.vector.rabitq import pack_nibbles

    rng = np.random.default_rng(9)
    for m, nq, C, dim in [(500, 3, 17, 64), (2049, 5, 333, 768),
                          (100, 2, 7, 99)]:
        codes = torch.from_numpy(rng.integers(0, 8, (m, dim)).astype(np.uint8))
        packed = pack_nibbles(codes).to(dev)
        q = torch.from_numpy(rng.normal(size=(nq, dim)).astype(np.float32)).to(dev)
        cand = torch.from_numpy(
            rng.integers(0, m, (nq, C)).astype(np.int64)).to(dev)
        cand[0, 0] = -1
        got = hip().fastscan_ex_dot_pairs(packed, cand, q, dim).cpu()
        dense = codes.to(torch.float32) @ q.cpu().T   # (m, nq)
        cc = cand.cpu()
        for qi in range(nq):
            for ci in range(C):
                r = int(cc[qi, ci])
                exp = 0.0 if r < 0 else float(dense[r, qi])
                assert abs(float(got[qi, ci]) - exp) <= 1e-2 + 1e-4 * abs(exp), \
                    (qi, ci, r)


def test_gpu_rabitq_index_recall(dev, tmp_path):
    """IVF-RaBitQ staged search on GPU (fastscan kernels + MFMA rescore):
    recall@10 vs exact on the same device."""
    from lakesoul_amd.meta.client import MetaClient
    from lakesoul_amd.meta.store import SqliteMetaStore
    from lakesoul_amd.tables.catalog import LakeSoulCatalog
    from lakesoul_amd.io.schema import Field, Schema
    from lakesoul_amd.vector.index import build_vector_index

    catalog = LakeSoulCatalog(
        MetaClient(SqliteMetaStore(str(tmp_path / "meta_rbq.db"))),
        warehouse=str(tmp_path / "wh_rbq"),
    )
    rng = np.random.default_rng(31)
    n, dim = 20000, 128
    t = catalog.create_table(
        "vecs_rbq",
        Schema([Field("id", "int64", False), Field("emb", "binary", False)]),
        primary_keys=["id"], hash_bucket_num=2,
    )
    vecs = rng.normal(size=(n, dim)).astype(np.float32)
    t.upsert({"id": np.arange(n, dtype=np.int64),
              "emb": [v.tobytes() for v in vecs]})
    exact = build_vector_index(t, "emb", metric="cosine", device="cuda")
    idx = build_vector_index(t, "emb", metric="cosine", rabitq_bits=4,
                             ivf_clusters=64, device="cuda")
    q = vecs[rng.choice(n, 32, replace=False)]
    ids_r, _ = idx.search(q, k=10, device="cuda", rescore=40)
    ids_e, _ = exact.search(q, k=10, device="cuda")
    recall = np.mean([len(set(ids_r[i]) & set(ids_e[i])) / 10.0
                      for i in range(32)])
    assert recall >= 0.85, recall
    assert (ids_r[:, 0] == ids_e[:, 0]).mean() >= 0.9


def test_ann_scores_t_matches_untransposed(dev):
    """Query-major MFMA output equals the row-major kernel transposed."""
    from lakesoul_amd.ops import hip

    rng = np.random.default_rng(11)
    for n, nq, k_ in [(1000, 16, 128), (4097, 32, 768)]:
        X = torch.from_numpy(rng.normal(size=(n, k_)).astype(np.float32)
                             ).to(torch.bfloat16).to(dev)
        Q = torch.from_numpy(rng.normal(size=(nq, k_)).astype(np.float32)
                             ).to(torch.bfloat16).to(dev)
        a = hip().ann_scores(X, Q)
        b = hip().ann_scores_t(X, Q)
        torch.testing.assert_close(b, a.T.contiguous())
