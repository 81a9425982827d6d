#!/usr/bin/env python3
"""Vector search quickstart: embeddings in a lakehouse table, exact MFMA
search, IVF pruning, and the 1-bit binary first pass.

    python examples/vector_search.py          # CPU works; MI355X is fast
"""

import os
import sys
import tempfile

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

workdir = tempfile.mkdtemp(prefix="lakesoul_vec_")
os.environ["LAKESOUL_META_DB"] = os.path.join(workdir, "meta.db")
os.environ["LAKESOUL_WAREHOUSE"] = os.path.join(workdir, "warehouse")

from lakesoul_amd import Field, LakeSoulCatalog, Schema          # noqa: E402
from lakesoul_amd.vector.index import build_vector_index         # noqa: E402

catalog = LakeSoulCatalog()
n, dim = 50_000, 128
rng = np.random.default_rng(0)
vecs = rng.normal(size=(n, dim)).astype(np.float32)

docs = catalog.create_table(
    "docs",
    Schema([Field("doc_id", "int64", False), Field("emb", "binary", False)]),
    primary_keys=["doc_id"],
    hash_bucket_num=4,
)
docs.upsert({"doc_id": np.arange(n, dtype=np.int64),
             "emb": [v.tobytes() for v in vecs]})

# exact bf16 MFMA search
idx = build_vector_index(docs, "emb", metric="cosine")
q = vecs[[7, 4242]]
ids, scores = idx.search(q, k=5)
print("exact top-5:", ids.tolist())
assert ids[0, 0] == 7 and ids[1, 0] == 4242

# IVF coarse quantizer (probe a subset of clusters)
ivf = build_vector_index(docs, "emb", metric="cosine", ivf_clusters=16)
ids_ivf, _ = ivf.search(q, k=5, nprobe=4)
print("ivf   top-5:", ids_ivf.tolist())

# 1-bit sign codes + rescore (16x less bandwidth on the first pass)
bin_idx = build_vector_index(docs, "emb", metric="cosine", binary=True)
ids_bin, _ = bin_idx.search(q, k=5, rescore=32)
print("1-bit top-5:", ids_bin.tolist())
assert ids_bin[0, 0] == 7

# IVF-RaBitQ (1 sign bit + 3 ex bits, the reference's quantizer) with
# staged search: fastscan estimate -> ex refine -> exact MFMA rescore
rbq = build_vector_index(docs, "emb", metric="cosine", rabitq_bits=4,
                         ivf_clusters=32)
ids_rbq, _ = rbq.search(q, k=5, rescore=40)
print("rabitq top-5:", ids_rbq.tolist())
assert ids_rbq[0, 0] == 7

# ANN results as a table scan filter (reader.rs:250-331 analog)
hits = docs.scan(columns=["doc_id"],
                 vector_query={"column": "emb", "query": vecs[7], "k": 5})
df = hits.to_arrow().to_pandas()
print("scan(vector_query) rows:", sorted(df["doc_id"].tolist()))
assert 7 in set(df["doc_id"].tolist())

print("vector search OK —", workdir)
