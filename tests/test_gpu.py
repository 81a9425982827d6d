"""GPU kernel + end-to-end tests (MI355X). Every test compares the HIP
path against the CPU oracle (numpy/pure-python murmur3 / merge_cpu)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available(), "GPU tests require an MI355X"
    return torch.device("cuda:0")


def test_hip_extension_is_native(dev):
    """The HIP extension must be loaded from the repo tree (native code check)."""
    import lakesoul_amd._hip as m

    assert m.__file__.endswith(".so")


def test_murmur3_fixed_gpu_matches_cpu(dev):
    from lakesoul_amd.ops import hip
    from lakesoul_amd.utils import murmur3 as m3

    rng = np.random.default_rng(0)
    empty_prev = torch.empty(0, dtype=torch.int64, device=dev)
    empty_val = torch.empty(0, dtype=torch.uint8, device=dev)
    for dt, name in [(np.int64, "int64"), (np.int32, "int32"),
                     (np.float32, "float32"), (np.float64, "float64")]:
        if dt in (np.int64, np.int32):
            vals = rng.integers(-10**9, 10**9, 1000).astype(dt)
        else:
            vals = rng.normal(size=1000).astype(dt)
        t = torch.from_numpy(vals).to(dev)
        h = hip().hash_fixed_column(t, empty_val, empty_prev, True).cpu().numpy()
        for i in range(0, 1000, 131):
            assert int(h[i]) == m3.hash_value(vals[i].item(), name), (name, i)


def test_murmur3_string_and_chaining_gpu(dev):
    from lakesoul_amd.ops import hip
    from lakesoul_amd.utils import murmur3 as m3

    strings = [f"key-{i}" for i in range(500)]
    enc = [s.encode() for s in strings]
    offs = np.zeros(501, dtype=np.int32)
    offs[1:] = np.cumsum([len(e) for e in enc])
    by = np.frombuffer(b"".join(enc), dtype=np.uint8).copy()
    empty_prev = torch.empty(0, dtype=torch.int64, device=dev)
    empty_val = torch.empty(0, dtype=torch.uint8, device=dev)
    h1 = hip().hash_string_column(
        torch.from_numpy(offs).to(dev), torch.from_numpy(by).to(dev),
        empty_val, empty_prev, True,
    )
    ids = torch.arange(500, dtype=torch.int64, device=dev)
    h2 = hip().hash_fixed_column(ids, empty_val, h1, False)
    got = h2.cpu().numpy()
    for i in range(0, 500, 61):
        expect = m3.hash_int64(i, m3.hash_str(strings[i]))
        assert int(got[i]) == expect
    b = hip().bucket_ids(h2, 16).cpu().numpy()
    for i in range(0, 500, 61):
        assert b[i] == m3.hash_int64(i, m3.hash_str(strings[i])) % 16


def test_merge_pairs_kernel(dev):
    from lakesoul_amd.ops import hip

    rng = np.random.default_rng(1)
    a = np.sort(rng.integers(0, 10**6, 100000).astype(np.uint64))
    b = np.sort(rng.integers(0, 10**6, 37777).astype(np.uint64))
    kA = torch.from_numpy(a.view(np.int64)).to(dev)
    kB = torch.from_numpy(b.view(np.int64)).to(dev)
    vA = torch.arange(len(a), dtype=torch.int64, device=dev)
    vB = torch.arange(1_000_000, 1_000_000 + len(b), dtype=torch.int64, device=dev)
    kO, vO = hip().merge_pairs(kA, vA, kB, vB)
    keys = kO.cpu().numpy().view(np.uint64)
    assert np.all(keys[1:] >= keys[:-1])
    # stability: for equal keys A values (< 1e6) come before B values
    vals = vO.cpu().numpy()
    eq = keys[1:] == keys[:-1]
    bad = eq & (vals[:-1] >= 1_000_000) & (vals[1:] < 1_000_000)
    assert not bad.any()
    # content preserved
    ref = np.sort(np.concatenate([a, b]), kind="stable")
    np.testing.assert_array_equal(keys, ref)


def test_rle_dict_decode_gpu_vs_pyarrow(dev, tmp_path):
    import pyarrow as pa
    import pyarrow.parquet as pq

    from lakesoul_amd.ops import cpp, hip

    n = 50000
    rng = np.random.default_rng(2)
    cat = (rng.integers(0, 100, n) * 7).astype(np.int64)
    vals = [None if rng.random() < 0.1 else int(cat[i]) for i in range(n)]
    tbl = pa.table({"c": pa.array(vals, type=pa.int64())})
    path = tmp_path / "dict.parquet"
    pq.write_table(tbl, str(path), use_dictionary=True, compression="zstd", row_group_size=20000)

    h = cpp().open_parquet(str(path))
    meta = cpp().parquet_meta(h)
    outs = []
    for rg in range(meta["num_row_groups"]):
        d = cpp().read_chunk_raw(h, rg, 0)
        assert d["is_dict"]
        from lakesoul_amd.io.reader_gpu import _decode_fixed_chunk_gpu

        col = _decode_fixed_chunk_gpu(d, "int64", dev)
        got = col.data.cpu().numpy()
        mask = col.validity.cpu().numpy() if col.validity is not None else np.ones(len(got), np.uint8)
        outs.append((got, mask))
    cpp().close_parquet(h)
    got = np.concatenate([o[0] for o in outs])
    mask = np.concatenate([o[1] for o in outs])
    for i in range(n):
        if vals[i] is None:
            assert mask[i] == 0
        else:
            assert mask[i] == 1 and got[i] == vals[i]


def test_gather_strings_kernel(dev):
    from lakesoul_amd.ops import hip

    strings = [f"string-value-{i:05d}" * (1 + i % 3) for i in range(1000)]
    enc = [s.encode() for s in strings]
    offs = np.zeros(1001, dtype=np.int64)
    offs[1:] = np.cumsum([len(e) for e in enc])
    by = torch.from_numpy(np.frombuffer(b"".join(enc), dtype=np.uint8).copy()).to(dev)
    offs_t = torch.from_numpy(offs).to(dev)
    idx = torch.from_numpy(np.random.default_rng(3).permutation(1000)[:300]).to(dev)
    lens = (offs_t[1:] - offs_t[:-1])[idx]
    new_offs = torch.zeros(301, dtype=torch.int64, device=dev)
    new_offs[1:] = torch.cumsum(lens, 0)
    out = hip().gather_strings(by, offs_t, idx, new_offs)
    ob = out.cpu().numpy().tobytes()
    no = new_offs.cpu().numpy()
    sel = idx.cpu().numpy()
    for i in range(300):
        assert ob[no[i]:no[i + 1]].decode() == strings[sel[i]]


def _mk_catalog(tmp_path):
    from lakesoul_amd.meta.client import MetaClient
    from lakesoul_amd.meta.store import SqliteMetaStore
    from lakesoul_amd.tables.catalog import LakeSoulCatalog

    store = SqliteMetaStore(str(tmp_path / "meta.db"))
    return LakeSoulCatalog(MetaClient(store), warehouse=str(tmp_path / "wh"))


def test_end_to_end_gpu_scan_matches_cpu(dev, tmp_path):
    from lakesoul_amd.io.schema import Field, Schema

    catalog = _mk_catalog(tmp_path)
    t = catalog.create_table(
        "gput",
        Schema([Field("id", "int64", False), Field("v", "float64"),
                Field("k", "int32"), Field("s", "string")]),
        primary_keys=["id"],
        hash_bucket_num=4,
    )
    n = 100000
    rng = np.random.default_rng(4)
    # write on GPU (GPU murmur3 + GPU sort)
    t.upsert(
        {
            "id": np.arange(n, dtype=np.int64),
            "v": rng.normal(size=n),
            "k": rng.integers(0, 100, n, dtype=np.int32),
            "s": [f"s{i}" for i in range(n)],
        },
        device="cuda",
    )
    for it in range(3):
        ids = rng.choice(n, 5000, replace=False).astype(np.int64)
        t.upsert(
            {
                "id": ids,
                "v": np.full(5000, float(it)),
                "k": np.full(5000, it, dtype=np.int32),
                "s": [f"u{it}"] * 5000,
            },
            device="cuda",
        )
    cpu_df = (
        t.scan(device="cpu").to_arrow().to_pandas().sort_values("id").reset_index(drop=True)
    )
    gpu_df = (
        t.scan(device="cuda").to_arrow().to_pandas().sort_values("id").reset_index(drop=True)
    )
    import pandas as pd

    pd.testing.assert_frame_equal(cpu_df, gpu_df)
    assert len(gpu_df) == n


def test_gpu_merge_operators_match_cpu(dev, tmp_path):
    from lakesoul_amd.io.schema import Field, Schema

    catalog = _mk_catalog(tmp_path)
    t = catalog.create_table(
        "gsum",
        Schema([Field("id", "int64", False), Field("cnt", "int64"), Field("x", "float64")]),
        primary_keys=["id"],
        hash_bucket_num=2,
        properties={"merge_op.cnt": "SumAll", "merge_op.x": "UseLastNotNull"},
    )
    rng = np.random.default_rng(5)
    for it in range(4):
        ids = rng.integers(0, 5000, 3000).astype(np.int64)
        ids = np.unique(ids)
        t.upsert(
            {
                "id": ids,
                "cnt": np.ones(len(ids), dtype=np.int64) * (it + 1),
                "x": rng.normal(size=len(ids)),
            },
            device="cpu",
        )
    import pandas as pd

    cpu_df = t.scan(device="cpu").to_arrow().to_pandas().sort_values("id").reset_index(drop=True)
    gpu_df = t.scan(device="cuda").to_arrow().to_pandas().sort_values("id").reset_index(drop=True)
    pd.testing.assert_frame_equal(cpu_df, gpu_df)


def test_gpu_cdc_delete(dev, tmp_path):
    from lakesoul_amd.io.schema import Field, Schema

    catalog = _mk_catalog(tmp_path)
    t = catalog.create_table(
        "gcdc",
        Schema([Field("id", "int64", False), Field("v", "float64"), Field("rowKinds", "string")]),
        primary_keys=["id"],
        properties={"lakesoul_cdc_change_column": "rowKinds"},
    )
    t.upsert({"id": np.arange(10, dtype=np.int64), "v": np.ones(10), "rowKinds": ["insert"] * 10})
    t.upsert({"id": np.array([3, 7], dtype=np.int64), "v": np.zeros(2), "rowKinds": ["delete", "delete"]})
    df = t.scan(device="cuda").to_arrow().to_pandas().sort_values("id")
    assert df["id"].tolist() == [0, 1, 2, 4, 5, 6, 8, 9]


def test_gpu_compaction_roundtrip(dev, tmp_path):
    from lakesoul_amd.io.schema import Field, Schema

    catalog = _mk_catalog(tmp_path)
    t = catalog.create_table(
        "gcomp",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"],
        hash_bucket_num=2,
    )
    n = 20000
    rng = np.random.default_rng(6)
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.zeros(n)}, device="cuda")
    for it in range(3):
        ids = rng.choice(n, 1000, replace=False).astype(np.int64)
        t.upsert({"id": ids, "v": np.full(1000, it + 1.0)}, device="cuda")
    before = t.scan(device="cuda").to_arrow().to_pandas().sort_values("id").reset_index(drop=True)
    t.compaction(device="cuda")
    after = t.scan(device="cuda").to_arrow().to_pandas().sort_values("id").reset_index(drop=True)
    import pandas as pd

    pd.testing.assert_frame_equal(before, after)
    assert all("compactdir" in f.path for f in t.files())


def test_gpu_filters_and_projection(dev, tmp_path):
    catalog = _mk_catalog(tmp_path)
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "gfilt",
        Schema([Field("id", "int64", False), Field("v", "float64"), Field("k", "int32")]),
        primary_keys=["id"],
        hash_bucket_num=4,
    )
    n = 50000
    t.upsert({"id": np.arange(n, dtype=np.int64),
              "v": np.arange(n, dtype=np.float64),
              "k": (np.arange(n) % 7).astype(np.int32)})
    # filter column (v) not in projection; DSL string form
    df = (
        t.scan(columns=["id"], filters="and(gteq(v, 100), lt(v, 200))", device="cuda")
        .to_arrow().to_pandas()
    )
    assert sorted(df["id"].tolist()) == list(range(100, 200))
    assert list(df.columns) == ["id"]
    # point lookup uses bucket pruning on GPU
    scan = t.scan(filters=[("id", "==", 4321)], device="cuda")
    assert len(scan.plan()) == 1
    df = scan.to_arrow().to_pandas()
    assert df["v"].tolist() == [4321.0]


def test_gpu_string_pk_hybrid(dev, tmp_path, monkeypatch):
    """String-PK tables scan on GPU via the hybrid CPU-merge path
    (LAKESOUL_GPU_STRING_MERGE=0 escape hatch)."""
    monkeypatch.setenv("LAKESOUL_GPU_STRING_MERGE", "0")
    catalog = _mk_catalog(tmp_path)
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "gspk",
        Schema([Field("key", "string", False), Field("v", "int64")]),
        primary_keys=["key"],
        hash_bucket_num=2,
    )
    t.upsert({"key": [f"customer_{i:05d}" for i in range(2000)],
              "v": np.arange(2000, dtype=np.int64)})
    t.upsert({"key": ["customer_00005", "zzz"], "v": np.array([-5, 1], dtype=np.int64)})
    df = t.scan(device="cuda").to_arrow().to_pandas().sort_values("key")
    assert len(df) == 2001
    got = dict(zip(df["key"], df["v"]))
    assert got["customer_00005"] == -5 and got["zzz"] == 1


def test_gpu_two_int32_pk_merge(dev, tmp_path):
    """Two-int32 composite PK uses the pack_key_2xi32 merge path."""
    catalog = _mk_catalog(tmp_path)
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "g2pk",
        Schema([Field("a", "int32", False), Field("b", "int32", False), Field("v", "float64")]),
        primary_keys=["a", "b"],
        hash_bucket_num=2,
    )
    rng = np.random.default_rng(7)
    a = rng.integers(-1000, 1000, 5000, dtype=np.int32)
    b = rng.integers(-1000, 1000, 5000, dtype=np.int32)
    t.upsert({"a": a, "b": b, "v": np.zeros(5000)})
    t.upsert({"a": a[:500], "b": b[:500], "v": np.ones(500)})
    import pandas as pd

    cpu = t.scan(device="cpu").to_arrow().to_pandas().sort_values(["a", "b"]).reset_index(drop=True)
    gpu = t.scan(device="cuda").to_arrow().to_pandas().sort_values(["a", "b"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(cpu, gpu)


def test_gpu_int32_single_pk(dev, tmp_path):
    catalog = _mk_catalog(tmp_path)
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "gi32",
        Schema([Field("id", "int32", False), Field("v", "float64")]),
        primary_keys=["id"],
        hash_bucket_num=2,
    )
    n = 30000
    t.upsert({"id": np.arange(n, dtype=np.int32), "v": np.zeros(n)})
    t.upsert({"id": np.arange(0, n, 7, dtype=np.int32), "v": np.ones(len(range(0, n, 7)))})
    df = t.scan(device="cuda").to_arrow().to_pandas().sort_values("id").reset_index(drop=True)
    assert len(df) == n
    assert (df["v"][::7] == 1.0).all() and (df["v"][1::7] == 0.0).all()


@pytest.mark.gpu
def test_gpu_decimal_mor(dev, tmp_path):
    """decimal column through the GPU MOR path (int64 unscaled on device)."""
    import decimal

    from lakesoul_amd.io.schema import Field, Schema

    gpu_catalog = _mk_catalog(tmp_path)
    t = gpu_catalog.create_table(
        "gdec",
        Schema([Field("id", "int64", False), Field("amt", "decimal(12,2)")]),
        primary_keys=["id"],
        hash_bucket_num=2,
    )
    n = 50000
    t.upsert({"id": np.arange(n, dtype=np.int64),
              "amt": (np.arange(n, dtype=np.int64) * 7) % 100000})
    t.upsert({"id": np.arange(0, n, 3, dtype=np.int64),
              "amt": np.full((n + 2) // 3, 123456, dtype=np.int64)})
    df = t.scan(device="cuda:0").to_arrow().to_pandas()
    df = df.sort_values("id").reset_index(drop=True)
    expect = (np.arange(n, dtype=np.int64) * 7) % 100000
    expect[::3] = 123456
    got = np.array([int(x.scaleb(2)) for x in df["amt"]])
    np.testing.assert_array_equal(got, expect)


@pytest.mark.gpu
def test_gpu_oversized_unit_fallback(dev, tmp_path, monkeypatch):
    """A bucket estimated over LAKESOUL_MAX_UNIT_BYTES must degrade
    gracefully (chunked PK-range merge for int PKs) and still return
    correct HBM-resident results."""
    import warnings

    from lakesoul_amd.io.schema import Field, Schema

    gpu_catalog = _mk_catalog(tmp_path)
    t = gpu_catalog.create_table(
        "big1",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"],
        hash_bucket_num=1,
    )
    n = 100000
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.zeros(n)})
    t.upsert({"id": np.arange(0, n, 2, dtype=np.int64), "v": np.ones(n // 2)})
    monkeypatch.setenv("LAKESOUL_MAX_UNIT_BYTES", "1000")
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        df = t.scan(device="cuda:0").to_arrow().to_pandas()
    assert any(("falling back" in str(x.message)) or ("chunked" in str(x.message))
               for x in w)
    df = df.sort_values("id").reset_index(drop=True)
    assert len(df) == n
    expect = np.zeros(n)
    expect[::2] = 1.0
    np.testing.assert_allclose(df["v"].to_numpy(), expect)


@pytest.mark.gpu
def test_gpu_zstd_kernel_matches_reference(dev):
    """Wave-per-page zstd kernel vs original bytes across content types
    (raw blocks, RLE, huffman literals, FSE sequences, multi-block)."""
    from lakesoul_amd.ops import cpp, hip

    rng = np.random.default_rng(7)
    payloads = []
    payloads.append(rng.integers(0, 256, 100000, dtype=np.uint8).tobytes())
    payloads.append(b"\x42" * 200000)
    payloads.append(np.arange(50000, dtype=np.int64).tobytes())
    payloads.append(rng.normal(size=40000).tobytes())
    payloads.append(bytes(rng.integers(65, 70, 150000, dtype=np.uint8)))
    words = [b"lake", b"soul", b"gpu", b"zstd", b"merge"]
    payloads.append(b" ".join(words[i] for i in rng.integers(0, 5, 80000)))
    payloads.append(b"x" * 300000)  # multi-block RLE
    for n in rng.integers(1, 5000, 40):
        payloads.append(bytes(rng.integers(0, 8, int(n), dtype=np.uint8)))

    for level in (1, 3, 9):
        comps = [cpp().zstd_compress_ref(p, level) for p in payloads]
        src = torch.from_numpy(
            np.frombuffer(b"".join(comps), dtype=np.uint8).copy()).to(dev)
        total_out = sum(len(p) for p in payloads)
        dst = torch.zeros(total_out, dtype=torch.uint8, device=dev)
        jobs = []
        so = do = 0
        for c, p in zip(comps, payloads):
            jobs.append([so, len(c), do, len(p)])
            so += len(c)
            do += len(p)
        jt = torch.tensor(jobs, dtype=torch.int64, device=dev)
        status = hip().zstd_decompress_into(src, jt, dst)
        torch.cuda.synchronize()
        assert int((status != 0).sum()) == 0, f"level {level}: {status.cpu().tolist()[:10]}"
        got = dst.cpu().numpy().tobytes()
        assert got == b"".join(payloads), f"level {level} content mismatch"


@pytest.mark.gpu
def test_gpu_zstd_scan_path_active(dev, tmp_path, monkeypatch):
    """End-to-end MOR scan with the GPU zstd path opted in: results equal
    the CPU scan, and the unit fetch reports zstd jobs (not host decode)."""
    import lakesoul_amd.io.reader_gpu as rgpu

    monkeypatch.setattr(rgpu, "_GPU_ZSTD", True)
    monkeypatch.setattr(rgpu, "_GPU_ZSTD_FRAC", 1.0)
    from lakesoul_amd.io.reader_gpu import fetch_raw

    catalog = _mk_catalog(tmp_path)
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "gz",
        Schema([Field("id", "int64", False), Field("v", "float64", False)]),
        primary_keys=["id"],
        hash_bucket_num=2,
    )
    n = 200000
    rng = np.random.default_rng(8)
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": rng.normal(size=n)})
    t.upsert({"id": np.arange(0, n, 3, dtype=np.int64),
              "v": np.full((n + 2) // 3, 7.5)})
    files = [f.path for f in t.files()]
    raw = fetch_raw(files, ["id", "v"])
    assert raw["zstd_jobs"].numel() > 0, "zstd pages should defer to the GPU"
    cpu_df = t.scan(device="cpu").to_arrow().to_pandas().sort_values("id").reset_index(drop=True)
    gpu_df = t.scan(device="cuda").to_arrow().to_pandas().sort_values("id").reset_index(drop=True)
    import pandas as pd

    pd.testing.assert_frame_equal(cpu_df, gpu_df)


@pytest.mark.gpu
def test_gpu_string_pk_native_merge(dev, tmp_path):
    """String-PK MOR merge fully on GPU (LSD chunk-key sort): results
    match the CPU merge on a multi-delta history with shared prefixes."""
    catalog = _mk_catalog(tmp_path)
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "gsnative",
        Schema([Field("key", "string", False), Field("v", "int64")]),
        primary_keys=["key"],
        hash_bucket_num=2,
    )
    n = 30000
    rng = np.random.default_rng(4)
    t.upsert({"key": [f"user_{i:08d}" for i in range(n)],
              "v": np.zeros(n, dtype=np.int64)})
    for it in range(4):
        ids = rng.choice(n, 5000, replace=False)
        t.upsert({"key": [f"user_{i:08d}" for i in ids],
                  "v": np.full(5000, it + 1, dtype=np.int64)})
    t.upsert({"key": ["a", "user_00000005\x00x", "zz"],
              "v": np.array([-1, -2, -3], dtype=np.int64)})
    cpu = t.scan(device="cpu").to_arrow().to_pandas().sort_values("key").reset_index(drop=True)
    gpu = t.scan(device="cuda").to_arrow().to_pandas().sort_values("key").reset_index(drop=True)
    import pandas as pd

    pd.testing.assert_frame_equal(cpu, gpu)
    assert len(gpu) == n + 3


@pytest.mark.gpu
def test_gpu_joined_merge_operators(dev, tmp_path):
    """JoinedAllByComma / JoinedLastBySemicolon on the GPU match the CPU
    oracle (delimiter-join per PK group, null poisoning)."""
    catalog = _mk_catalog(tmp_path)
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "gjoin",
        Schema([Field("id", "int64", False), Field("tags", "string"),
                Field("last", "string")]),
        primary_keys=["id"],
        hash_bucket_num=2,
        properties={"merge_op.tags": "JoinedAllByComma",
                    "merge_op.last": "JoinedLastBySemicolon"},
    )
    n = 5000
    t.upsert({"id": np.arange(n, dtype=np.int64),
              "tags": [f"a{i}" for i in range(n)],
              "last": [f"x{i}" for i in range(n)]})
    rng = np.random.default_rng(6)
    for it in range(3):
        ids = np.sort(rng.choice(n, 800, replace=False)).astype(np.int64)
        t.upsert({"id": ids,
                  "tags": [f"b{it}_{i}" for i in ids],
                  "last": [f"y{it}_{i}" for i in ids]})
    # a null value poisons its group
    import pyarrow as pa

    t.upsert(pa.table({
        "id": pa.array([0, 1], pa.int64()),
        "tags": pa.array([None, "z"], pa.string()),
        "last": pa.array(["q", None], pa.string()),
    }))
    cpu = t.scan(device="cpu").to_arrow().to_pandas().sort_values("id").reset_index(drop=True)
    gpu = t.scan(device="cuda").to_arrow().to_pandas().sort_values("id").reset_index(drop=True)
    import pandas as pd

    pd.testing.assert_frame_equal(cpu, gpu)
    assert cpu["tags"].iloc[0] is None or pd.isna(cpu["tags"].iloc[0])
    assert cpu["tags"].str.contains(",").fillna(False).any()
    assert cpu["last"].str.contains(";").fillna(False).any()


@pytest.mark.gpu
def test_gpu_range_partitioned_mor(dev, tmp_path):
    """Range partitions (col=val dirs) + hash buckets + GPU MOR merge +
    partition pruning, results equal CPU."""
    catalog = _mk_catalog(tmp_path)
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "grange",
        Schema([Field("dt", "string", False), Field("id", "int64", False),
                Field("v", "float64", False)]),
        primary_keys=["id"],
        range_partitions=["dt"],
        hash_bucket_num=2,
    )
    n = 40000
    for day in ("2026-01-01", "2026-01-02"):
        t.upsert({"dt": [day] * n, "id": np.arange(n, dtype=np.int64),
                  "v": np.zeros(n)})
        t.upsert({"dt": [day] * (n // 4),
                  "id": np.arange(0, n, 4, dtype=np.int64),
                  "v": np.ones(n // 4)})
    cpu = t.scan(device="cpu", partitions=["dt=2026-01-02"]).to_arrow().to_pandas()
    gpu = t.scan(device="cuda", partitions=["dt=2026-01-02"]).to_arrow().to_pandas()
    cpu = cpu.sort_values("id").reset_index(drop=True)
    gpu = gpu.sort_values("id").reset_index(drop=True)
    import pandas as pd

    pd.testing.assert_frame_equal(cpu, gpu)
    assert len(gpu) == n and (gpu["dt"] == "2026-01-02").all()
    expect = np.zeros(n)
    expect[::4] = 1.0
    np.testing.assert_allclose(gpu["v"].to_numpy(), expect)


def test_gpu_list_column_mor(dev, tmp_path):
    """list<float32> columns scan through the GPU unit path (element
    payload in the string lane, whole-value UseLast merge on device):
    equal to the CPU scan, incl. null lists and updates."""
    catalog = _mk_catalog(tmp_path)
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "gl",
        Schema([Field("id", "int64", False), Field("emb", "list<float32>"),
                Field("v", "float64")]),
        primary_keys=["id"], hash_bucket_num=2,
    )
    n = 5000
    base = [None if i % 11 == 0 else
            list(np.float32(i) + np.arange(i % 4, dtype=np.float32))
            for i in range(n)]
    t.upsert({"id": np.arange(n, dtype=np.int64), "emb": base,
              "v": np.arange(n, dtype=np.float64)})
    t.upsert({"id": np.array([3, 44], dtype=np.int64),
              "emb": [[9.0], None], "v": np.array([3.5, 44.5])})
    cpu = t.scan(device="cpu").to_arrow().to_pandas().sort_values("id").reset_index(drop=True)
    gpu = t.scan(device="cuda").to_arrow().to_pandas().sort_values("id").reset_index(drop=True)
    assert len(gpu) == n
    import pandas as pd

    for i in range(n):
        a, b = cpu["emb"].iloc[i], gpu["emb"].iloc[i]
        a_null = a is None or (isinstance(a, float) and np.isnan(a))
        b_null = b is None or (isinstance(b, float) and np.isnan(b))
        assert a_null == b_null, i
        if not a_null:
            np.testing.assert_allclose(list(b), list(a), err_msg=str(i))
    np.testing.assert_allclose(gpu["v"].to_numpy(), cpu["v"].to_numpy())
    np.testing.assert_allclose(list(gpu["emb"].iloc[3]), [9.0])


def test_gpu_list_string_scan(dev, tmp_path):
    """list<string> scans on device='cuda' route through the host
    decode+merge gate and ship the parsed column to HBM; values match
    the CPU scan and the tensors land on the GPU."""
    catalog = _mk_catalog(tmp_path)
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "gls",
        Schema([Field("id", "int64", False), Field("tags", "list<string>"),
                Field("v", "float64")]),
        primary_keys=["id"], hash_bucket_num=2,
    )
    n = 2000
    base = [None if i % 11 == 0 else
            [f"t{i}", ""] [: i % 3] for i in range(n)]
    t.upsert({"id": np.arange(n, dtype=np.int64), "tags": base,
              "v": np.arange(n, dtype=np.float64)})
    t.upsert({"id": np.array([3, 44], dtype=np.int64),
              "tags": [["zz"], None], "v": np.array([3.5, 44.5])})
    scan = t.scan(device="cuda")
    for b in scan.iter_batches():
        c = b.columns["tags"]
        assert c.offsets.device.type == "cuda"
        assert c.elem_offsets.device.type == "cuda"
    cpu = t.scan(device="cpu").to_arrow().sort_by("id")
    gpu = t.scan(device="cuda").to_arrow().sort_by("id")
    assert gpu.column("tags").to_pylist() == cpu.column("tags").to_pylist()
    assert gpu.column("tags").to_pylist()[3] == ["zz"]


def test_gpu_struct_map_scan(dev, tmp_path):
    """struct/map columns on device='cuda' host-decode and ship to HBM;
    values match the CPU scan and child tensors land on the GPU."""
    catalog = _mk_catalog(tmp_path)
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "gsm",
        Schema([Field("id", "int64", False),
                Field("st", "struct<a:int64,b:string>"),
                Field("m", "map<string,int64>")]),
        primary_keys=["id"], hash_bucket_num=2,
    )
    n = 500
    st0 = [{"a": i, "b": f"s{i}"} if i % 5 else None for i in range(n)]
    m0 = [None if i % 7 == 0 else
          {f"k{j}": i * 10 + j for j in range(i % 3)} for i in range(n)]
    t.upsert({"id": np.arange(n, dtype=np.int64), "st": st0, "m": m0})
    t.upsert({"id": np.array([1, 3], dtype=np.int64),
              "st": [{"a": 100, "b": "upd"}, None],
              "m": [{"z": 9}, {"y": 8}]})
    for b in t.scan(device="cuda").iter_batches():
        assert b.columns["st"].children["a"].data.device.type == "cuda"
        assert b.columns["m"].children["key"].offsets.device.type == "cuda"
    cpu = t.scan(device="cpu").to_arrow().sort_by("id")
    gpu = t.scan(device="cuda").to_arrow().sort_by("id")
    assert gpu.column("st").to_pylist() == cpu.column("st").to_pylist()
    assert gpu.column("m").to_pylist() == cpu.column("m").to_pylist()
    assert gpu.column("st").to_pylist()[1] == {"a": 100, "b": "upd"}


def test_gpu_struct_native_unit_path(dev, tmp_path):
    """struct-of-scalars (and map of primitives) scans stay on the GPU
    unit path — leaves decode as flat/list device columns and reassemble
    at projection (no host decode gate)."""
    catalog = _mk_catalog(tmp_path)
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "gsn",
        Schema([Field("id", "int64", False),
                Field("st", "struct<a:int64,b:string>"),
                Field("mi", "map<int32,float64>")]),
        primary_keys=["id"], hash_bucket_num=2,
    )
    n = 4000
    st0 = [{"a": i * 2, "b": f"s{i}"} if i % 5 else None for i in range(n)]
    mi0 = [None if i % 7 == 0 else
           {j: float(i + j) for j in range(i % 3)} for i in range(n)]
    t.upsert({"id": np.arange(n, dtype=np.int64), "st": st0, "mi": mi0})
    t.upsert({"id": np.array([3, 44], dtype=np.int64),
              "st": [{"a": -1, "b": "upd"}, None],
              "mi": [{9: 9.5}, None]})
    scan = t.scan(device="cuda")
    assert not scan._has_list_str()  # native path engaged
    for b in scan.iter_batches():
        assert b.columns["st"].children["a"].data.device.type == "cuda"
        assert b.columns["st"].children["b"].offsets.device.type == "cuda"
        assert b.columns["mi"].children["key"].offsets.device.type == "cuda"
    cpu = t.scan(device="cpu").to_arrow().sort_by("id")
    gpu = t.scan(device="cuda").to_arrow().sort_by("id")
    assert gpu.column("st").to_pylist() == cpu.column("st").to_pylist()
    assert gpu.column("mi").to_pylist() == cpu.column("mi").to_pylist()
    assert gpu.column("st").to_pylist()[3] == {"a": -1, "b": "upd"}


def test_gpu_oversized_string_pk_chunked(dev, tmp_path, monkeypatch):
    """String-PK oversized buckets chunk-merge by lexicographic PK
    ranges and ship each part to HBM — same rows as the CPU scan."""
    import warnings

    from lakesoul_amd.io.schema import Field, Schema

    gpu_catalog = _mk_catalog(tmp_path)
    t = gpu_catalog.create_table(
        "bigstr",
        Schema([Field("k", "string", False), Field("v", "float64")]),
        primary_keys=["k"], hash_bucket_num=1,
    )
    n = 50000
    keys = [f"x{i:06d}" for i in range(n)]
    t.upsert({"k": keys, "v": np.zeros(n)})
    t.upsert({"k": keys[::2], "v": np.ones(n // 2)})
    full = t.scan(device="cpu").to_arrow().to_pandas().sort_values("k").reset_index(drop=True)
    monkeypatch.setenv("LAKESOUL_MAX_UNIT_BYTES", "1000")
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        df = t.scan(device="cuda").to_arrow().to_pandas().sort_values("k").reset_index(drop=True)
    assert any("chunked" in str(x.message) for x in w), \
        [str(x.message) for x in w]
    import pandas as pd

    pd.testing.assert_frame_equal(full, df)
