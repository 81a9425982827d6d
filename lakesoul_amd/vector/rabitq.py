"""RaBitQ quantization + estimator — the reference's lakesoul-vector
core (rust/lakesoul-vector/src/rabitq/quantizer.rs, ivf/lut.rs,
fastscan_kernel.rs) re-derived as vectorized tensor math.

Encoding (per vector v, against its IVF cluster centroid c, in rotated
space): residual r = v - c; 1 sign bit per dim plus ``ex_bits`` magnitude
bits per dim (ex codes of negative dims are bit-complemented, so
total_code = ex + (bit << ex_bits) is a monotone staircase of r). Per
vector correction factors (f_add, f_rescale, f_error and their _ex
variants, plus delta/vl for reconstruction) make the integer inner
product <total_code, q> an unbiased distance estimator:

  1-bit:  est  = f_add   + g_add + f_rescale   * (<bits, q> + c1*sum_q)
  lower   bound = est - f_error * g_error
  ex-ref: dist = f_add_ex + g_add + f_rescale_ex *
                 (2^eb * <bits, q> + <ex, q> + cb*sum_q)

with per-(query, cluster) constants g_add = ||q-c||^2 (L2) or -<q,c>
(IP), g_error = ||q-c||, c1 = -0.5, cb = -(2^eb - 0.5). The estimator
consumes the FULL query (not q - c): centroid cross terms live in the
per-vector factors — this is what lets one LUT serve every cluster
(reference lut.hpp analog, ivf/lut.rs:92).

All ops are batched over the vector axis (torch), so the same code runs
on CPU (test oracle) and GPU; the HIP FastScan kernel consumes the
packed arrays this module produces.
"""

from __future__ import annotations

import heapq
import math
from dataclasses import dataclass
from typing import Optional, Tuple

import numpy as np
import torch

K_TIGHT_START = [0.0, 0.15, 0.20, 0.52, 0.59, 0.71, 0.75, 0.77, 0.81]
K_EPS = 1e-5
K_NENUM = 10.0
K_CONST_EPSILON = 1.9


def best_rescale_factor(o_abs: np.ndarray, ex_bits: int) -> float:
    """Optimal per-vector rescale factor t (reference quantizer.rs:319-397:
    incremental sweep over the breakpoints of floor(t*|o|))."""
    dim = len(o_abs)
    max_o = float(o_abs.max()) if dim else 0.0
    if max_o <= np.finfo(np.float64).eps:
        return 1.0
    ti = min(ex_bits, len(K_TIGHT_START) - 1)
    t_end = ((1 << ex_bits) - 1 + K_NENUM) / max_o
    t_start = t_end * K_TIGHT_START[ti]

    cur = (t_start * o_abs + K_EPS).astype(np.int64)
    sqr_den = dim * 0.25 + float((cur * cur + cur).sum())
    num = float(((cur + 0.5) * o_abs).sum())

    heap = []
    for i in range(dim):
        if o_abs[i] > 0:
            heapq.heappush(heap, ((cur[i] + 1) / o_abs[i], i))
    max_ip, best_t = 0.0, t_start
    maxv = (1 << ex_bits) - 1
    while heap:
        t, i = heapq.heappop(heap)
        if t >= t_end:
            continue
        cur[i] += 1
        sqr_den += 2.0 * cur[i]
        num += float(o_abs[i])
        ip = num / math.sqrt(sqr_den)
        if ip > max_ip:
            max_ip, best_t = ip, t
        if cur[i] < maxv and o_abs[i] > 0:
            nt = (cur[i] + 1) / o_abs[i]
            if nt < t_end:
                heapq.heappush(heap, (nt, i))
    return best_t if best_t > 0 else max(t_start, np.finfo(np.float64).eps)


def compute_const_scaling_factor(dim: int, ex_bits: int, seed: int = 0,
                                 samples: int = 100) -> float:
    """Average optimal t over random gaussian directions (reference
    quantizer.rs:546-574 'faster' config: <1% accuracy loss)."""
    rng = np.random.default_rng(seed)
    acc = 0.0
    n = 0
    for _ in range(samples):
        v = rng.normal(size=dim).astype(np.float32)
        nrm = np.linalg.norm(v)
        if nrm <= np.finfo(np.float32).eps:
            continue
        acc += best_rescale_factor(np.abs(v / nrm), ex_bits)
        n += 1
    return float(acc / max(n, 1))


@dataclass
class QuantizedBatch:
    """Per-cluster quantized vectors + factors (struct-of-arrays)."""
    dim: int
    ex_bits: int
    bits_packed: torch.Tensor    # (n, ceil(dim/8)) uint8, LSB-first
    ex_packed: torch.Tensor      # (n, ceil(dim/2)) uint8 nibbles (lo=even dim) or empty
    f_add: torch.Tensor          # (n,) f32
    f_rescale: torch.Tensor
    f_error: torch.Tensor
    f_add_ex: torch.Tensor
    f_rescale_ex: torch.Tensor
    delta: torch.Tensor
    vl: torch.Tensor

    @property
    def n(self) -> int:
        return self.bits_packed.shape[0]

    def to(self, device) -> "QuantizedBatch":
        return QuantizedBatch(
            self.dim, self.ex_bits,
            *(getattr(self, f).to(device) for f in (
                "bits_packed", "ex_packed", "f_add", "f_rescale", "f_error",
                "f_add_ex", "f_rescale_ex", "delta", "vl")))


def pack_bits(bits: torch.Tensor) -> torch.Tensor:
    """(n, dim) bool -> (n, ceil(dim/8)) uint8 LSB-first."""
    n, dim = bits.shape
    pad = (-dim) % 8
    if pad:
        bits = torch.cat([bits, torch.zeros(n, pad, dtype=bits.dtype,
                                            device=bits.device)], dim=1)
    b = bits.view(n, -1, 8).to(torch.uint8)
    w = (1 << torch.arange(8, dtype=torch.uint8, device=bits.device))
    return (b * w).sum(dim=2, dtype=torch.int64).to(torch.uint8)


def unpack_bits(packed: torch.Tensor, dim: int) -> torch.Tensor:
    """(n, ceil(dim/8)) uint8 -> (n, dim) uint8 {0,1}."""
    n = packed.shape[0]
    shifts = torch.arange(8, device=packed.device, dtype=torch.uint8)
    u = (packed.unsqueeze(2) >> shifts) & 1
    return u.view(n, -1)[:, :dim]


def pack_nibbles(codes: torch.Tensor) -> torch.Tensor:
    """(n, dim) uint8 (<16) -> (n, ceil(dim/2)) uint8; even dim in the
    low nibble."""
    n, dim = codes.shape
    pad = dim % 2
    if pad:
        codes = torch.cat([codes, torch.zeros(n, 1, dtype=codes.dtype,
                                              device=codes.device)], dim=1)
    c = codes.view(n, -1, 2)
    return (c[:, :, 0] | (c[:, :, 1] << 4)).to(torch.uint8)


def unpack_nibbles(packed: torch.Tensor, dim: int) -> torch.Tensor:
    n = packed.shape[0]
    lo = packed & 0x0F
    hi = (packed >> 4) & 0x0F
    out = torch.stack([lo, hi], dim=2).view(n, -1)
    return out[:, :dim]


def quantize_batch(vecs: torch.Tensor, centroid: torch.Tensor, ex_bits: int,
                   t_const: Optional[float], metric: str) -> QuantizedBatch:
    """Quantize (n, dim) f32 vectors against one centroid. metric: 'l2'
    or 'ip' (cosine uses 'ip' on normalized inputs). Mirrors
    quantize_with_centroid (quantizer.rs:113-233) batched over n."""
    assert metric in ("l2", "ip")
    v = vecs.to(torch.float32)
    c = centroid.to(torch.float32)
    n, dim = v.shape
    r = v - c
    bits = (r >= 0)
    eb = ex_bits
    maxv = (1 << eb) - 1 if eb else 0

    l2_sqr = (r * r).sum(1)
    l2_norm = l2_sqr.sqrt()
    eps = torch.finfo(torch.float32).eps

    if eb > 0:
        if t_const is None:
            t_const = compute_const_scaling_factor(dim, eb)
        o_abs = r.abs() / l2_norm.clamp_min(eps).unsqueeze(1)
        o_abs = torch.where(l2_norm.unsqueeze(1) > eps, o_abs,
                            torch.zeros_like(o_abs))
        cur = (t_const * o_abs + K_EPS).floor().clamp(0, maxv).to(torch.int32)
        ipnorm = ((cur.to(torch.float32) + 0.5) * o_abs).sum(1).to(torch.float64)
        ipnorm_inv = torch.where(
            (ipnorm > 0) & torch.isfinite(ipnorm), 1.0 / ipnorm,
            torch.ones_like(ipnorm)).to(torch.float32)
        ex_code = torch.where(r < 0, (~cur) & maxv, cur).to(torch.int32)
    else:
        ipnorm_inv = torch.ones(n, dtype=torch.float32, device=v.device)
        ex_code = torch.zeros(n, dim, dtype=torch.int32, device=v.device)

    total = ex_code + (bits.to(torch.int32) << eb)

    # ---- 1-bit factors (compute_one_bit_factors) ---- #
    xu_cb = bits.to(torch.float32) - 0.5
    xu_cb_norm_sqr = torch.full_like(l2_sqr, dim * 0.25)
    ip_resi_xucb = (r * xu_cb).sum(1)
    ip_cent_xucb = xu_cb @ c
    dot_r_c = r @ c

    denom = torch.where(ip_resi_xucb.abs() <= eps,
                        torch.full_like(ip_resi_xucb, float("inf")),
                        ip_resi_xucb)
    tmp_error = torch.zeros_like(l2_sqr)
    if dim > 1:
        ratio = (l2_sqr * xu_cb_norm_sqr) / (denom * denom) - 1.0
        ok = torch.isfinite(ratio) & (ratio > 0)
        tmp_error = torch.where(
            ok,
            l2_norm * K_CONST_EPSILON * (ratio.clamp_min(0) / (dim - 1)).sqrt(),
            tmp_error)
    if metric == "l2":
        f_add = l2_sqr + 2.0 * l2_sqr * ip_cent_xucb / denom
        f_rescale = -2.0 * l2_sqr / denom
        f_error = 2.0 * tmp_error
    else:
        f_add = 1.0 - dot_r_c + l2_sqr * ip_cent_xucb / denom
        f_rescale = -l2_sqr / denom
        f_error = tmp_error

    # ---- delta / vl (reconstruction) ---- #
    cb = -((1 << eb) - 0.5)
    qs = total.to(torch.float32) + cb
    norm_quan = (qs * qs).sum(1).sqrt()
    dot_r_q = (r * qs).sum(1)
    denom2 = (l2_norm * norm_quan).clamp_min(eps)
    cos_sim = (dot_r_q / denom2).clamp(-1.0, 1.0)
    delta = torch.where(norm_quan <= eps, torch.zeros_like(cos_sim),
                        l2_norm / norm_quan.clamp_min(eps) * cos_sim)
    vl = delta * cb

    # ---- extended factors (compute_extended_factors) ---- #
    if eb > 0:
        ip_resi_xucb_ex = (r * qs).sum(1)
        ip_cent_xucb_ex = qs @ c
        safe_denom = torch.where(ip_resi_xucb_ex.abs() <= eps,
                                 torch.full_like(ip_resi_xucb_ex, float("inf")),
                                 ip_resi_xucb_ex)
        if metric == "l2":
            f_add_ex = l2_sqr + 2.0 * l2_sqr * ip_cent_xucb_ex / safe_denom
            f_rescale_ex = -2.0 * l2_norm * ipnorm_inv
        else:
            f_add_ex = 1.0 - dot_r_c + l2_sqr * ip_cent_xucb_ex / safe_denom
            f_rescale_ex = -l2_norm * ipnorm_inv
    else:
        f_add_ex = torch.zeros_like(f_add)
        f_rescale_ex = torch.zeros_like(f_add)

    return QuantizedBatch(
        dim=dim, ex_bits=eb,
        bits_packed=pack_bits(bits),
        ex_packed=(pack_nibbles(ex_code.to(torch.uint8)) if eb
                   else torch.empty(n, 0, dtype=torch.uint8, device=v.device)),
        f_add=f_add.to(torch.float32),
        f_rescale=f_rescale.to(torch.float32),
        f_error=f_error.to(torch.float32),
        f_add_ex=f_add_ex.to(torch.float32),
        f_rescale_ex=f_rescale_ex.to(torch.float32),
        delta=delta.to(torch.float32),
        vl=vl.to(torch.float32),
    )


def reconstruct(qb: QuantizedBatch, centroid: torch.Tensor) -> torch.Tensor:
    """centroid + delta*total_code + vl (quantizer.rs reconstruct_into)."""
    bits = unpack_bits(qb.bits_packed, qb.dim).to(torch.int32)
    ex = (unpack_nibbles(qb.ex_packed, qb.dim).to(torch.int32) if qb.ex_bits
          else torch.zeros_like(bits))
    total = (ex + (bits << qb.ex_bits)).to(torch.float32)
    return centroid.unsqueeze(0) + qb.delta.unsqueeze(1) * total + qb.vl.unsqueeze(1)


@dataclass
class QueryContext:
    """Per-query precompute (ivf/lut.rs QueryPrecomputed)."""
    q: torch.Tensor        # (dim,) f32 rotated query
    sum_q: float
    norm_q: float
    ex_bits: int

    @property
    def c1_sum_q(self) -> float:
        return -0.5 * self.sum_q

    @property
    def cb_sum_q(self) -> float:
        return -((1 << self.ex_bits) - 0.5) * self.sum_q

    @property
    def binary_scale(self) -> float:
        return float(1 << self.ex_bits)


def make_query(q: torch.Tensor, ex_bits: int) -> QueryContext:
    q = q.to(torch.float32)
    return QueryContext(q, float(q.sum()), float(q.norm()), ex_bits)


def cluster_g(qc: QueryContext, centroid: torch.Tensor, metric: str
              ) -> Tuple[float, float]:
    """(g_add, g_error) per cluster (ivf/mod.rs:1285-1293)."""
    d = qc.q - centroid.to(torch.float32)
    cd = float((d * d).sum())
    if metric == "l2":
        return cd, math.sqrt(cd)
    return -float(qc.q @ centroid.to(torch.float32)), math.sqrt(cd)


def estimate_1bit(qb: QuantizedBatch, qc: QueryContext, g_add: float,
                  g_error: float) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """(ip_x0_qr, est_distance, lower_bound) for every vector in the
    batch — the FastScan stage-1 estimator (fastscan_kernel.rs:24-96),
    computed exactly (float dot; the HIP kernel uses the LDS-LUT
    approximation of the same quantity)."""
    bits = unpack_bits(qb.bits_packed, qb.dim).to(torch.float32)
    ip = bits @ qc.q.to(bits.device)
    est = qb.f_add + g_add + qb.f_rescale * (ip + qc.c1_sum_q)
    lb = est - qb.f_error * g_error
    return ip, est, lb


def refine_ex(qb: QuantizedBatch, qc: QueryContext, idx: torch.Tensor,
              ip_x0_qr: torch.Tensor, g_add: float) -> torch.Tensor:
    """Stage-2 refinement with ex codes for the selected rows ``idx``
    (fastscan_kernel.rs:130-160 refine_distance_with_ex)."""
    if qb.ex_bits == 0:
        raise ValueError("no ex bits to refine with")
    ex = unpack_nibbles(qb.ex_packed[idx], qb.dim).to(torch.float32)
    ex_dot = ex @ qc.q.to(ex.device)
    total_term = qc.binary_scale * ip_x0_qr[idx] + ex_dot + qc.cb_sum_q
    return qb.f_add_ex[idx] + g_add + qb.f_rescale_ex[idx] * total_term
