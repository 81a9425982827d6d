"""Vector ANN index over table columns.

MI355X-native analog of the reference's ``rust/lakesoul-vector`` (IVF +
RaBitQ with AVX kernels) + the glue in ``rust/lakesoul-io/src/vector/``
(builder.rs:32-121, search.rs:15-162): per-hash-bucket index shards
stored as sidecar files under ``<table_path>/_vector_index/<column>/``
with a JSON manifest.

Round-1 engine: exact search on MFMA matrix cores — vectors stored bf16
(L2-normalized for cosine), scored with the hand-written
mfma_f32_16x16x32_bf16 kernel (csrc/hip/ann.hip) at HBM speed, per-shard
top-k merged across shards. 288 GB HBM keeps even billion-row 768-d
tables resident. An IVF coarse quantizer (GPU k-means) can narrow the
candidate set; exact scan is the round-1 default (the quality bar the
reference's RaBitQ approximates).

The fixed-point list-member layout (one shard per hash bucket, row ids
are positions in the bucket's merged scan order) matches the reference's
bucket-keyed shard scheme (search.rs:115).
"""

from __future__ import annotations

import json
import os
import time
from dataclasses import dataclass
from typing import List, Optional, Tuple

import numpy as np
import torch


@dataclass
class ShardInfo:
    bucket_id: int
    num_rows: int
    path: str


def random_rotation(dim: int, seed: int = 7) -> np.ndarray:
    """Seeded random orthonormal rotation (QR of a gaussian) — isotropizes
    coordinates before 1-bit sign quantization (the reference's RaBitQ
    rotation, rabitq/rotation.rs)."""
    rng = np.random.default_rng(seed)
    a = rng.normal(size=(dim, dim)).astype(np.float64)
    q_m, r = np.linalg.qr(a)
    q_m *= np.sign(np.diag(r))  # deterministic sign convention
    return q_m.astype(np.float32)


def pack_sign_bits(x: np.ndarray) -> np.ndarray:
    """(n, dim) float -> (n, ceil(dim/64)) int64 sign codes (bit i of
    word w = sign(x[:, 64*w + i]) >= 0)."""
    n, dim = x.shape
    bits = (x >= 0).astype(np.uint8)
    pad = (-dim) % 64
    if pad:
        bits = np.concatenate([bits, np.zeros((n, pad), np.uint8)], axis=1)
    packed = np.packbits(bits, axis=1, bitorder="little")
    return packed.view(np.int64).reshape(n, -1)


def kmeans(x: torch.Tensor, k: int, iters: int = 10, seed: int = 0) -> torch.Tensor:
    """Plain Lloyd k-means on the active device (the reference trains its
    IVF coarse quantizer the same way, rabitq/kmeans.rs). Returns
    L2-normalized centroids (cosine geometry)."""
    g = torch.Generator(device="cpu").manual_seed(seed)
    init = torch.randperm(x.shape[0], generator=g)[:k]
    c = x[init.to(x.device)].clone()
    for _ in range(iters):
        scores = x @ c.T  # cosine (inputs normalized)
        assign = scores.argmax(dim=1)
        for j in range(k):
            m = assign == j
            if bool(m.any()):
                c[j] = x[m].mean(dim=0)
        c = c / c.norm(dim=1, keepdim=True).clamp_min(1e-30)
    return c


class VectorIndex:
    def __init__(self, root: str, column: str, dim: int, metric: str,
                 shards: List[ShardInfo], pk_dtype: str, version: int,
                 ivf_clusters: int = 0, binary: bool = False,
                 rabitq_bits: int = 0):
        self.root = root
        self.column = column
        self.dim = dim
        self.metric = metric
        self.shards = shards
        self.pk_dtype = pk_dtype
        self.version = version
        self.ivf_clusters = ivf_clusters
        self.binary = binary
        self.rabitq_bits = rabitq_bits
        self._gpu_cache: dict = {}

    # -- persistence ---------------------------------------------------- #

    def manifest_path(self) -> str:
        return os.path.join(self.root, "manifest.json")

    def save_manifest(self) -> None:
        m = {
            "column": self.column,
            "dim": self.dim,
            "metric": self.metric,
            "pk_dtype": self.pk_dtype,
            "version": self.version,
            "created_ms": int(time.time() * 1000),
            "engine": ("ivf-rabitq+mfma-rescore" if self.rabitq_bits
                       else "binary+mfma-rescore" if self.binary
                       else "mfma-exact-bf16" if not self.ivf_clusters
                       else "ivf+mfma-bf16"),
            "ivf_clusters": self.ivf_clusters,
            "binary": self.binary,
            "rabitq_bits": self.rabitq_bits,
            "shards": [
                {"bucket_id": s.bucket_id, "num_rows": s.num_rows, "path": s.path}
                for s in self.shards
            ],
        }
        os.makedirs(self.root, exist_ok=True)
        with open(self.manifest_path(), "w") as f:
            json.dump(m, f, indent=1)

    @classmethod
    def load(cls, root: str) -> "VectorIndex":
        with open(os.path.join(root, "manifest.json")) as f:
            m = json.load(f)
        shards = [ShardInfo(s["bucket_id"], s["num_rows"], s["path"]) for s in m["shards"]]
        return cls(root, m["column"], m["dim"], m["metric"], shards,
                   m["pk_dtype"], m["version"], m.get("ivf_clusters", 0),
                   m.get("binary", False), m.get("rabitq_bits", 0))

    # -- shard data ----------------------------------------------------- #

    def _load_shard(self, s: ShardInfo, device):
        key = (s.path, str(device))
        if key in self._gpu_cache:
            return self._gpu_cache[key]
        raw = np.fromfile(s.path + ".vec", dtype=np.uint16).reshape(s.num_rows, self.dim)
        vecs = torch.from_numpy(raw.view(np.int16)).view(torch.bfloat16).to(device)
        ids = torch.from_numpy(np.fromfile(s.path + ".ids", dtype=np.int64)).to(device)
        clu = None
        if self.ivf_clusters and os.path.exists(s.path + ".clu"):
            clu = torch.from_numpy(np.fromfile(s.path + ".clu", dtype=np.int64))
        codes = None
        if self.binary:
            w = (self.dim + 63) // 64
            codes = torch.from_numpy(
                np.fromfile(s.path + ".bin", dtype=np.int64).reshape(s.num_rows, w)
            ).to(device)
        rbq = None
        if self.rabitq_bits:
            from .rabitq import QuantizedBatch

            eb = self.rabitq_bits - 1
            n = s.num_rows
            bits = torch.from_numpy(np.fromfile(
                s.path + ".rbq.bits", dtype=np.uint8).reshape(n, (self.dim + 7) // 8))
            if eb:
                ex = torch.from_numpy(np.fromfile(
                    s.path + ".rbq.ex", dtype=np.uint8).reshape(n, (self.dim + 1) // 2))
            else:
                ex = torch.empty(n, 0, dtype=torch.uint8)
            fac = torch.from_numpy(np.fromfile(
                s.path + ".rbq.fac", dtype=np.float32).reshape(7, n))
            rbq = QuantizedBatch(self.dim, eb, bits, ex, fac[0], fac[1], fac[2],
                                 fac[3], fac[4], fac[5], fac[6]).to(device)
        entry = (vecs, ids, clu, codes, rbq)
        self._gpu_cache[key] = entry
        return entry

    def _rotation(self) -> Optional[np.ndarray]:
        if not self.binary:
            return None
        key = "__rotation__"
        if key not in self._gpu_cache:
            self._gpu_cache[key] = np.fromfile(
                os.path.join(self.root, "rotation.vec"), dtype=np.float32
            ).reshape(self.dim, self.dim)
        return self._gpu_cache[key]

    def _centroids(self, device) -> Optional[torch.Tensor]:
        if not self.ivf_clusters:
            return None
        key = ("__centroids__", str(device))
        if key not in self._gpu_cache:
            raw = np.fromfile(os.path.join(self.root, "centroids.vec"),
                              dtype=np.float32).reshape(self.ivf_clusters, self.dim)
            self._gpu_cache[key] = torch.from_numpy(raw).to(device)
        return self._gpu_cache[key]

    # -- search --------------------------------------------------------- #

    def search(self, queries, k: int = 10, device: Optional[str] = None,
               nprobe: Optional[int] = None, rescore: int = 16):
        """Top-k over all shards. Returns (ids, scores) arrays of shape
        (nq, k). Cosine: inputs are normalized; score = cosine similarity.
        L2: score = -||x-q||^2 (larger is better).

        With an IVF index, only the members of the ``nprobe`` nearest
        coarse clusters per query are scored (default: clusters/8, min 4
        — the reference's IVF SearchParams analog)."""
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        q = torch.as_tensor(np.asarray(queries, dtype=np.float32))
        if q.dim() == 1:
            q = q[None, :]
        nq = q.shape[0]
        if self.metric == "cosine":
            q = q / q.norm(dim=1, keepdim=True).clamp_min(1e-30)
        q_dev = q.to(device)

        if self.rabitq_bits:
            # dispatch BEFORE nprobe defaulting: the fastscan engine's
            # default is a full scan (probing is a mask there, not a
            # traffic saver)
            return self._search_rabitq(q_dev, k, device, nprobe, rescore)
        best_scores = torch.full((nq, k), -float("inf"), device=device)
        best_ids = torch.full((nq, k), -1, dtype=torch.int64, device=device)
        cents = self._centroids(device)
        if cents is not None and nprobe is None:
            nprobe = max(4, self.ivf_clusters // 8)
        qcodes = None
        if self.binary:
            rot = self._rotation()
            qr = q.numpy() @ rot
            qcodes = torch.from_numpy(pack_sign_bits(qr)).to(device)
        for s in self.shards:
            vecs, ids, clu, codes, _rbq = self._load_shard(s, device)
            if cents is not None and clu is not None:
                # union of the nprobe nearest clusters over the query batch
                cscores = q_dev.to(cents.dtype) @ cents.T  # (nq, k_c)
                probe = torch.topk(cscores, min(nprobe, self.ivf_clusters), dim=1).indices
                wanted = torch.unique(probe.flatten()).cpu()
                # vectors are stored cluster-sorted: gather member ranges
                segs = []
                for cid in wanted.tolist():
                    a, b = int(clu[cid]), int(clu[cid + 1])
                    if b > a:
                        segs.append((a, b))
                if not segs:
                    continue
                vecs = torch.cat([vecs[a:b] for a, b in segs])
                ids = torch.cat([ids[a:b] for a, b in segs])
                if codes is not None:
                    codes = torch.cat([codes[a:b] for a, b in segs])
            if codes is not None:
                # 1-bit first pass: hamming distance on sign codes, then
                # exact MFMA rescore of rescore*k candidates per query
                nc = min(max(k * rescore, k), codes.shape[0])
                if str(device).startswith("cuda"):
                    from ..ops import hip

                    ham = hip().hamming_scores(codes, qcodes)  # (n, nq) i32
                else:
                    x = codes.numpy().view(np.uint64)
                    qq = qcodes.numpy().view(np.uint64)
                    ham = torch.from_numpy(
                        np.bitwise_count(x[:, None, :] ^ qq[None, :, :])
                        .sum(axis=2).astype(np.int32))
                # transpose first: topk over the strided dim of (n, nq)
                # costs 11x (benchmarks/topk_micro.py)
                cand = torch.topk(-ham.T.contiguous().to(torch.float32),
                                  nc, dim=1).indices            # (nq, nc)
                flat = torch.unique(cand.flatten())
                sub_t = self._scores_t(vecs[flat], q_dev, device)  # (nq, m)
                scores_t = torch.full((nq, vecs.shape[0]), -float("inf"),
                                      device=sub_t.device)
                scores_t[:, flat] = sub_t
            else:
                scores_t = self._scores_t(vecs, q_dev, device)  # (nq, n) f32
            kk = min(k, scores_t.shape[1])
            top = torch.topk(scores_t, kk, dim=1)  # (nq, kk)
            cand_scores = torch.cat([best_scores, top.values], dim=1)
            cand_ids = torch.cat([best_ids, ids[top.indices]], dim=1)
            sel = torch.topk(cand_scores, k, dim=1)
            best_scores = sel.values
            best_ids = torch.gather(cand_ids, 1, sel.indices)
        return best_ids.cpu().numpy(), best_scores.cpu().numpy()

    def _search_rabitq(self, q_dev: torch.Tensor, k: int, device,
                       nprobe: Optional[int], rescore: int):
        """IVF + RaBitQ staged search (reference ivf/mod.rs
        search_cluster_v2_batched): probe -> 1-bit FastScan estimate ->
        ex-code refine -> exact bf16 MFMA rescore -> top-k.

        On GPU the 1-bit pass runs as the LDS-LUT fastscan HIP kernel
        when available; the torch path computes the same estimate
        exactly (it is the CPU oracle the kernel is tested against)."""
        from .rabitq import make_query, unpack_bits, unpack_nibbles

        nq = q_dev.shape[0]
        metric = "ip" if self.metric == "cosine" else "l2"
        eb = self.rabitq_bits - 1
        cents = self._centroids(device)
        if cents is None:
            raise ValueError("rabitq index requires IVF centroids")
        if nprobe is None:
            # the 1-bit pass scans every row in one kernel (96 B/vec);
            # probe masking prunes candidates, not traffic — default to
            # no pruning for best recall at equal cost
            nprobe = self.ivf_clusters
        nprobe = min(nprobe, self.ivf_clusters)
        C = max(128 * k, rescore * k)    # stage-1 candidates per query
        R = max(4 * k, rescore)          # exact-rescore budget per query

        sum_q = q_dev.sum(dim=1)                      # (nq,)
        c1_sum_q = -0.5 * sum_q
        cb = -((1 << eb) - 0.5)
        cb_sum_q = cb * sum_q
        bscale = float(1 << eb)

        from ..utils import timing as _tm

        best_scores = torch.full((nq, k), -float("inf"), device=device)
        best_ids = torch.full((nq, k), -1, dtype=torch.int64, device=device)
        for s in self.shards:
            vecs, ids, clu, _codes, rbq = self._load_shard(s, device)
            n = s.num_rows
            # row -> cluster (cached per shard/device)
            ckey = (s.path, str(device), "cl_of_row")
            if ckey not in self._gpu_cache:
                counts = (clu[1:] - clu[:-1]).to(torch.int64)
                self._gpu_cache[ckey] = torch.repeat_interleave(
                    torch.arange(len(counts), dtype=torch.int64),
                    counts).to(device)
            cl_of_row = self._gpu_cache[ckey]
            # per-(query, cluster) g_add
            if metric == "l2":
                qn = (q_dev * q_dev).sum(1, keepdim=True)
                cn = (cents * cents).sum(1)[None, :]
                g_add_all = qn - 2.0 * (q_dev @ cents.T) + cn
            else:
                g_add_all = -(q_dev @ cents.T)

            # stage 1: ONE fastscan pass over the whole shard (96 B/vec of
            # HBM traffic at 768-d — scanning everything beats per-cluster
            # launches; probe pruning is a mask, not a loop). On GPU the
            # fused kernel applies the correction factors in-register and
            # writes est directly (no (nq, n) intermediate round trips).
            use_hip = str(device).startswith("cuda")
            ip_T = None
            if use_hip:
                try:
                    from ..ops import hip

                    with _tm.phase("vq_fastscan_est", sync_gpu=True):
                        est = hip().fastscan_est(
                            rbq.bits_packed, q_dev, self.dim, rbq.f_add,
                            rbq.f_rescale, cl_of_row.to(torch.int32),
                            g_add_all.contiguous(), c1_sum_q.contiguous(),
                        )                                       # (nq, n)
                except (ImportError, AttributeError, RuntimeError):
                    use_hip = False
            if not use_hip:
                bits_f = unpack_bits(rbq.bits_packed, self.dim).to(torch.float32)
                ip = bits_f @ q_dev.T                           # (n, nq)
                ip_T = ip.T
                est = (rbq.f_add[None, :] + g_add_all[:, cl_of_row]
                       + rbq.f_rescale[None, :] * (ip_T + c1_sum_q[:, None]))
            if nprobe < self.ivf_clusters:
                cscores = q_dev.to(cents.dtype) @ cents.T       # (nq, kc)
                probe = torch.topk(cscores, nprobe, dim=1).indices
                probe_mask = torch.zeros(nq, cents.shape[0], dtype=torch.bool,
                                         device=device)
                probe_mask.scatter_(1, probe, True)
                est = torch.where(probe_mask[:, cl_of_row], est,
                                  torch.full_like(est, float("inf")))
            kk = min(C, n)
            with _tm.phase("vq_stage1_topk", sync_gpu=True):
                top_c = torch.topk(-est, kk, dim=1)
            cand_est = -top_c.values                            # (nq, kk)
            cand_row = top_c.indices
            if ip_T is not None:
                cand_ip = torch.gather(ip_T, 1, cand_row)
            else:
                # recover <bits, q> algebraically on the candidates only
                fr = rbq.f_rescale[cand_row]
                ga = torch.gather(g_add_all, 1, cl_of_row[cand_row])
                cand_ip = torch.where(
                    fr != 0,
                    (cand_est - rbq.f_add[cand_row] - ga) / fr
                    - c1_sum_q[:, None],
                    torch.zeros_like(cand_est))

            # stage 2: ex-code refinement of the C candidates
            _ex_t = _tm.phase("vq_ex_refine", sync_gpu=True)
            _ex_t.__enter__()
            if eb > 0:
                refined = torch.full_like(cand_est, float("inf"))
                valid = cand_row >= 0
                rows = cand_row.clamp_min(0)
                use_hip_ex = str(device).startswith("cuda")
                ex_dot = None
                if use_hip_ex:
                    try:
                        from ..ops import hip

                        # pair-wise kernel: one dot per (query, candidate)
                        # — the unique-rows variant recomputed each row's
                        # dot against EVERY query (nq-x waste, 75% of the
                        # staged search; profiles/r02_vector_recall_qps.md)
                        ex_dot = hip().fastscan_ex_dot_pairs(
                            rbq.ex_packed, rows, q_dev, self.dim)  # (nq, C)
                    except (ImportError, AttributeError, RuntimeError):
                        use_hip_ex = False
                if ex_dot is None:
                    flat_rows = torch.unique(rows.flatten())
                    ex_rows = rbq.ex_packed[flat_rows]
                    ex_f = unpack_nibbles(ex_rows, self.dim).to(torch.float32)
                    ex_dots_flat = ex_f @ q_dev.T               # (u, nq)
                    pos = torch.searchsorted(flat_rows, rows.flatten()
                                             ).view(rows.shape)
                    ex_dot = torch.gather(ex_dots_flat.T, 1, pos)  # (nq, C)
                # per-candidate cluster id for g_add
                g_add_cand = torch.gather(g_add_all, 1, cl_of_row[rows])
                total_term = (bscale * cand_ip + ex_dot + cb_sum_q[:, None])
                refined = (rbq.f_add_ex[rows] + g_add_cand
                           + rbq.f_rescale_ex[rows] * total_term)
                refined = torch.where(valid, refined,
                                      torch.full_like(refined, float("inf")))
            else:
                refined = cand_est
            _ex_t.__exit__(None, None, None)

            rr = min(R, refined.shape[1])
            _rs_t = _tm.phase("vq_exact_rescore", sync_gpu=True)
            _rs_t.__enter__()
            top_r = torch.topk(-refined, rr, dim=1).indices     # (nq, rr)
            rescore_rows = torch.gather(cand_row, 1, top_r)
            valid_r = rescore_rows >= 0
            flat = torch.unique(rescore_rows.clamp_min(0).flatten())
            sub_t = self._scores_t(vecs[flat], q_dev, device)   # (nq, u) exact
            posr = torch.searchsorted(flat, rescore_rows.clamp_min(0).flatten()
                                      ).view(rescore_rows.shape)
            exact = torch.gather(sub_t, 1, posr)
            exact = torch.where(valid_r, exact,
                                torch.full_like(exact, -float("inf")))
            shard_ids = ids[rescore_rows.clamp_min(0)]
            cand_scores = torch.cat([best_scores, exact], dim=1)
            cand_ids = torch.cat([best_ids, shard_ids], dim=1)
            sel = torch.topk(cand_scores, k, dim=1)
            best_scores = sel.values
            best_ids = torch.gather(cand_ids, 1, sel.indices)
            _rs_t.__exit__(None, None, None)
        return best_ids.cpu().numpy(), best_scores.cpu().numpy()

    def _scores_t(self, vecs: torch.Tensor, q: torch.Tensor, device) -> torch.Tensor:
        """Query-major exact scores (nq, n): the MFMA kernel writes the
        transposed layout directly so per-query top-k reads contiguous
        rows (torch.topk over the strided dim of (n, nq) costs 11x —
        benchmarks/topk_micro.py)."""
        nq = q.shape[0]
        if str(device).startswith("cuda"):
            from ..ops import hip

            nq_pad = (nq + 15) // 16 * 16
            qb = torch.zeros(nq_pad, self.dim, dtype=torch.bfloat16, device=device)
            qb[:nq] = q.to(torch.bfloat16)
            if self.metric == "cosine":
                return hip().ann_scores_t(vecs, qb)[:nq]
            dots = hip().ann_scores_t(vecs, qb)[:nq]
            xn = vecs.to(torch.float32).pow(2).sum(1)
            qn = q.pow(2).sum(1)
            return 2 * dots - xn[None, :] - qn[:, None]
        return self._scores(vecs, q, device).T.contiguous()

    def _scores(self, vecs: torch.Tensor, q: torch.Tensor, device) -> torch.Tensor:
        n = vecs.shape[0]
        nq = q.shape[0]
        if str(device).startswith("cuda"):
            from ..ops import hip

            nq_pad = (nq + 15) // 16 * 16
            qb = torch.zeros(nq_pad, self.dim, dtype=torch.bfloat16, device=device)
            qb[:nq] = q.to(torch.bfloat16)
            if self.metric == "cosine":
                s = hip().ann_scores(vecs, qb)[:, :nq]
            else:  # l2: -|x|^2 + 2 x.q - |q|^2
                dots = hip().ann_scores(vecs, qb)[:, :nq]
                xn = vecs.to(torch.float32).pow(2).sum(1, keepdim=True)
                qn = q.pow(2).sum(1)[None, :]
                s = 2 * dots - xn - qn
            return s
        # CPU fallback (API tests without GPU)
        xv = vecs.to(torch.float32)
        if self.metric == "cosine":
            return xv @ q.T
        dots = xv @ q.T
        xn = xv.pow(2).sum(1, keepdim=True)
        qn = q.pow(2).sum(1)[None, :]
        return 2 * dots - xn - qn


def build_vector_index(
    table,
    column: str,
    pk: Optional[str] = None,
    metric: str = "cosine",
    device: Optional[str] = None,
    ivf_clusters: int = 0,
    binary: bool = False,
    rabitq_bits: int = 0,
) -> VectorIndex:
    """Build per-bucket exact-search shards for a fixed-size-list float
    column stored as ``dim`` float32/float64 scalar columns or via numpy
    packing. The table must have integer PKs (stored as row ids).

    Vector columns: we store vectors in the table as ``binary`` cells of
    dim*4 bytes (little-endian f32) — same physical idea as the
    reference's Arrow FixedSizeList binary layout.
    """
    if pk is None:
        pks = table.primary_keys
        if len(pks) != 1:
            raise ValueError("vector index needs a single integer PK")
        pk = pks[0]
    root = os.path.join(table.table_path, "_vector_index", column)
    os.makedirs(root, exist_ok=True)
    shards: List[ShardInfo] = []
    _shard_payloads: list = []
    dim = None
    scan = table.scan(columns=[pk, column], device=device or "cpu")
    for unit in scan.plan():
        batch = scan._read_unit(unit)
        if batch is None or batch.num_rows == 0:
            continue
        c = batch.columns[column]
        ids_t = batch.columns[pk].data
        offs = c.offsets.cpu().numpy()
        raw = c.bytes_.cpu().numpy().tobytes()
        n = batch.num_rows
        if dim is None:
            dim = (offs[1] - offs[0]) // 4
        vecs = np.frombuffer(raw, dtype=np.float32).reshape(n, dim).copy()
        if metric == "cosine":
            norms = np.linalg.norm(vecs, axis=1, keepdims=True)
            norms[norms == 0] = 1
            vecs = vecs / norms
        spath = os.path.join(root, f"shard_{unit.bucket_id:04d}")
        _shard_payloads.append((spath, unit.bucket_id, vecs, ids_t.cpu().numpy().astype(np.int64)))
    if dim is None:
        raise ValueError("no data to index")

    dev = device or ("cuda" if torch.cuda.is_available() else "cpu")
    if rabitq_bits:
        # RaBitQ needs per-cluster centroids (reference: quantization is
        # always relative to the IVF centroid, ivf/builder.rs)
        if not ivf_clusters:
            n_total = sum(len(p[2]) for p in _shard_payloads)
            ivf_clusters = int(min(4096, max(8, n_total // 4096)))
        if binary:
            raise ValueError("choose either binary or rabitq_bits, not both")
    rot = None
    if binary:
        rot = random_rotation(dim)
        rot.tofile(os.path.join(root, "rotation.vec"))
    centroids = None
    if ivf_clusters:
        # coarse quantizer on a sample (GPU k-means; rabitq/kmeans.rs analog)
        sample = np.concatenate([p[2] for p in _shard_payloads])
        if len(sample) > 200_000:
            sel = np.random.default_rng(0).choice(len(sample), 200_000, replace=False)
            sample = sample[sel]
        centroids = kmeans(torch.from_numpy(sample).to(dev), ivf_clusters).cpu()
        centroids.numpy().astype(np.float32).tofile(os.path.join(root, "centroids.vec"))

    for spath, bucket_id, vecs, ids_np in _shard_payloads:
        n = len(vecs)
        order = np.arange(n)
        if centroids is not None:
            assign = (torch.from_numpy(vecs).to(dev) @ centroids.to(dev).T).argmax(dim=1).cpu().numpy()
            order = np.argsort(assign, kind="stable")
            counts = np.bincount(assign, minlength=ivf_clusters)
            clu = np.zeros(ivf_clusters + 1, dtype=np.int64)
            np.cumsum(counts, out=clu[1:])
            clu.tofile(spath + ".clu")
        v_sorted = vecs[order]
        vbf = torch.from_numpy(v_sorted).to(torch.bfloat16).view(torch.int16).numpy().view(np.uint16)
        vbf.tofile(spath + ".vec")
        ids_np[order].tofile(spath + ".ids")
        if rot is not None:
            pack_sign_bits(v_sorted @ rot).tofile(spath + ".bin")
        if rabitq_bits:
            from .rabitq import compute_const_scaling_factor, quantize_batch

            eb = rabitq_bits - 1
            if "_rbq_t" not in locals():
                _rbq_t = (compute_const_scaling_factor(dim, eb) if eb else None)
            qmetric = "ip" if metric == "cosine" else "l2"
            vt = torch.from_numpy(v_sorted).to(dev)
            ct = centroids.to(dev)
            bits_parts, ex_parts, fac_parts = [], [], []
            for cid in range(ivf_clusters):
                a, b = int(clu[cid]), int(clu[cid + 1])
                if b <= a:
                    continue
                qb = quantize_batch(vt[a:b], ct[cid], eb, _rbq_t, qmetric)
                bits_parts.append(qb.bits_packed.cpu())
                ex_parts.append(qb.ex_packed.cpu())
                fac_parts.append(torch.stack([
                    qb.f_add, qb.f_rescale, qb.f_error, qb.f_add_ex,
                    qb.f_rescale_ex, qb.delta, qb.vl]).cpu())
            torch.cat(bits_parts).numpy().tofile(spath + ".rbq.bits")
            if eb:
                torch.cat(ex_parts).numpy().tofile(spath + ".rbq.ex")
            torch.cat(fac_parts, dim=1).numpy().astype(np.float32).tofile(
                spath + ".rbq.fac")
        shards.append(ShardInfo(bucket_id, n, spath))

    version = table.latest_version() or 0
    idx = VectorIndex(root, column, int(dim), metric, shards, "int64", version,
                      ivf_clusters=ivf_clusters, binary=binary,
                      rabitq_bits=rabitq_bits)
    idx.save_manifest()
    return idx


def vector_search(table, column: str, queries, k: int = 10, device=None):
    root = os.path.join(table.table_path, "_vector_index", column)
    idx = VectorIndex.load(root)
    return idx.search(queries, k, device=device)
