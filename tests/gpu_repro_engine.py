"""Staged GPU repro for the query-engine arrow corruption — run directly
on a GPU box: python tests/gpu_repro_engine.py"""


import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

from lakesoul_amd.io.batch import Batch
from lakesoul_amd.io.schema import Field, Schema
from lakesoul_amd.query.engine import (
    factorize, groupby_agg, hash_join, join_batches, sort_indices)

rng = np.random.default_rng(0)
n = 200_000
schema = Schema([Field("g", "string"), Field("k", "int64", False),
                 Field("v", "float64")])
b = Batch.from_dict({
    "g": [f"grp{i % 37}" for i in range(n)],
    "k": rng.integers(0, 1000, n),
    "v": rng.normal(size=n),
}, schema).to_device("cuda")

print("== stage 1: take(permutation) on gpu strings ==")
perm = torch.randperm(n, device="cuda")
t1 = b.take(perm)
offs = t1.columns["g"].offsets.cpu()
print("offsets monotone:", bool((offs[1:] >= offs[:-1]).all()),
      "last:", int(offs[-1]), "bytes:", int(t1.columns["g"].bytes_.numel()))
assert int(offs[-1]) == int(t1.columns["g"].bytes_.numel()), "take corrupt"
_ = t1.to_device("cpu").to_arrow()
print("take->arrow OK")

print("== stage 2: factorize strings on gpu ==")
codes, g, rep = factorize([b.columns["g"]])
print("g:", g, "rep max:", int(rep.max()), "codes max:", int(codes.max()))
assert g == 37

print("== stage 3: groupby ==")
out = groupby_agg(b, ["g"], [("count", None, "n", False),
                             ("sum", "v", "sv", False),
                             ("max", "k", "mk", False)])
df = out.to_arrow().to_pandas()
print("groupby->arrow OK,", len(df), "groups, n sum:", df["n"].sum())
assert df["n"].sum() == n

print("== stage 4: join ==")
rschema = Schema([Field("k2", "int64", False), Field("w", "float64")])
r = Batch.from_dict({"k2": np.arange(1000, dtype=np.int64),
                     "w": np.ones(1000)}, rschema).to_device("cuda")
li, ri = hash_join(b, r, ["k"], ["k2"], "inner")
print("join rows:", li.numel(), "li range:", int(li.min()), int(li.max()),
      "ri range:", int(ri.min()), int(ri.max()))
assert int(li.max()) < n and int(ri.max()) < 1000 and int(li.min()) >= 0
joined = join_batches(b, r, ["k"], ["k2"], "inner")
joffs = joined.columns["g"].offsets.cpu()
print("joined g offsets last:", int(joffs[-1]),
      "bytes:", int(joined.columns["g"].bytes_.numel()))
assert int(joffs[-1]) == int(joined.columns["g"].bytes_.numel()), "join gather corrupt"
_ = joined.slice(0, 1000).to_device("cpu").to_arrow()
print("join->arrow OK")

print("== stage 5: string sort ==")
idx = sort_indices(b, [("g", True), ("v", False)])
su = torch.unique(idx)
print("idx is permutation:", su.numel() == n, int(idx.min()), int(idx.max()))
assert su.numel() == n
t5 = b.take(idx)
offs5 = t5.columns["g"].offsets.cpu()
assert int(offs5[-1]) == int(t5.columns["g"].bytes_.numel()), "sorted take corrupt"
got = t5.to_device("cpu").to_arrow().to_pandas()
assert got["g"].tolist() == sorted(got["g"].tolist()), "sort order wrong"
print("sort->arrow OK")
print("ALL OK")
