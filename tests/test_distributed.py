"""Multi-process distributed tests (gloo, world_size=2, CPU).

These cover the collective shard-exchange paths that run over RCCL on
the 8-GPU box; gloo exercises the identical code on CPU.
"""

import os
import socket

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _init(rank, world, rdv_file):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    import torch.distributed as dist

    # file rendezvous: immune to TCP port reuse races across test runs
    dist.init_process_group(
        "gloo", rank=rank, world_size=world, init_method=f"file://{rdv_file}"
    )
    return dist


def _run_exchange(rank, world, port, tmpdir, results):
    dist = _init(rank, world, port)
    try:
        from lakesoul_amd.io.batch import Batch
        from lakesoul_amd.io.schema import Field, Schema
        from lakesoul_amd.parallel.shard import exchange_batch_all_to_all

        n = 100
        ids = np.arange(rank * 1000, rank * 1000 + n, dtype=np.int64)
        strings = [f"r{rank}-{i}" for i in range(n)]
        schema = Schema([Field("id", "int64", False), Field("s", "string")])
        batch = Batch.from_dict({"id": ids, "s": strings}, schema)
        dest = torch.from_numpy((ids % world).astype(np.int64))
        out = exchange_batch_all_to_all(batch, dest)
        got_ids = out.columns["id"].data.numpy()
        # every received id must satisfy id % world == rank
        assert (got_ids % world == rank).all()
        # strings came along consistently
        offs = out.columns["s"].offsets.numpy()
        bys = out.columns["s"].bytes_.numpy().tobytes()
        for i, gid in enumerate(got_ids):
            src_rank = gid // 1000
            assert bys[offs[i]:offs[i + 1]].decode() == f"r{src_rank}-{gid % 1000}"
        # total rows conserved
        t = torch.tensor([out.num_rows])
        dist.all_reduce(t)
        assert int(t.item()) == world * n
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()


def _run_table_shard(rank, world, port, tmpdir, results):
    dist = _init(rank, world, port)
    try:
        os.environ["LAKESOUL_META_DB"] = os.path.join(tmpdir, "meta.db")
        from lakesoul_amd.meta.client import MetaClient
        from lakesoul_amd.meta.store import SqliteMetaStore
        from lakesoul_amd.tables.catalog import LakeSoulCatalog
        from lakesoul_amd.io.schema import Field, Schema
        from lakesoul_amd.torch.dataset import LakeSoulIterableDataset

        catalog = LakeSoulCatalog(
            MetaClient(SqliteMetaStore(os.environ["LAKESOUL_META_DB"])),
            warehouse=os.path.join(tmpdir, "wh"),
        )
        if rank == 0:
            t = catalog.create_table(
                "dshard",
                Schema([Field("id", "int64", False), Field("v", "float64")]),
                primary_keys=["id"],
                hash_bucket_num=8,
            )
            n = 8000
            t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.zeros(n)})
            t.upsert({"id": np.arange(0, n, 3, dtype=np.int64), "v": np.ones(len(range(0, n, 3)))})
        dist.barrier()
        t = catalog.table("dshard")
        ds = LakeSoulIterableDataset(t, device="cpu")
        ids = []
        for item in ds:
            ids.append(item["id"].numpy())
        my_ids = np.concatenate(ids) if ids else np.empty(0, np.int64)
        # disjoint cover across ranks
        lens = torch.tensor([len(my_ids)])
        dist.all_reduce(lens)
        assert int(lens.item()) == 8000
        # concurrent commit from both ranks (MVCC under multi-process)
        t.upsert({"id": np.array([100000 + rank], dtype=np.int64), "v": np.array([1.0])})
        dist.barrier()
        assert t.scan(device="cpu").count() == 8002
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()


def _run_rebalance(rank, world, rdv_file, tmpdir, results):
    dist = _init(rank, world, rdv_file)
    try:
        from lakesoul_amd.io.batch import Batch
        from lakesoul_amd.io.schema import Field, Schema
        from lakesoul_amd.parallel.shard import rebalance_by_pk
        from lakesoul_amd.utils.murmur3_np import bucket_ids_np, hash_column

        n = 500
        ids = np.arange(rank * 10000, rank * 10000 + n, dtype=np.int64)
        schema = Schema([Field("id", "int64", False), Field("v", "float64")])
        batch = Batch.from_dict({"id": ids, "v": ids.astype(np.float64)}, schema)
        out = rebalance_by_pk(batch, "id")
        got = out.columns["id"].data.numpy()
        # every received id murmur-hashes to this rank
        h = hash_column(got, np.uint32(42))
        assert (bucket_ids_np(h, world) == rank).all()
        # payload stayed aligned
        np.testing.assert_allclose(out.columns["v"].data.numpy(),
                                   got.astype(np.float64))
        t = torch.tensor([out.num_rows])
        dist.all_reduce(t)
        assert int(t.item()) == world * n
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()


def _run_exchange_hard(rank, world, rdv_file, tmpdir, results):
    """The exact collective shapes bench.py uses at world=4/8: nullable
    columns, strings, validity, skewed destinations (some peers receive
    nothing), one entirely-empty rank, async overlap path."""
    dist = _init(rank, world, rdv_file)
    try:
        from lakesoul_amd.io.batch import Batch
        from lakesoul_amd.io.schema import Field, Schema
        from lakesoul_amd.parallel.shard import (
            exchange_batch_all_to_all, exchange_batch_all_to_all_async)

        schema = Schema([
            Field("id", "int64", False),
            Field("v", "float64", True),
            Field("k", "int32", False),
            Field("s", "string", True),
        ])

        # case 1: skewed — every row goes to rank 0 (peers >0 receive 0)
        n = 64 if rank != world - 1 else 0  # last rank sends nothing at all
        ids = np.arange(rank * 1000, rank * 1000 + n, dtype=np.int64)
        batch = Batch.from_dict({
            "id": ids,
            "v": ids.astype(np.float64),
            "k": (ids % 7).astype(np.int32),
            "s": [None if i % 5 == 0 else f"r{rank}:{i}" for i in range(n)],
        }, schema)
        if batch.columns["v"].validity is None and n:
            batch.columns["v"].validity = torch.tensor(
                [0 if i % 3 == 0 else 1 for i in range(n)], dtype=torch.uint8)
        dest = torch.zeros(n, dtype=torch.int64)
        out = exchange_batch_all_to_all(batch, dest)
        t = torch.tensor([out.num_rows])
        dist.all_reduce(t)
        expect_total = 64 * (world - 1)
        assert int(t.item()) == expect_total
        if rank == 0:
            assert out.num_rows == expect_total
            got = out.columns["id"].data.numpy()
            offs = out.columns["s"].offsets.numpy()
            bys = out.columns["s"].bytes_.numpy().tobytes()
            sval = out.columns["s"].validity
            vval = out.columns["v"].validity.numpy()
            for i, gid in enumerate(got):
                src, idx = gid // 1000, gid % 1000
                if idx % 5 == 0:
                    assert sval is None or sval.numpy()[i] == 0
                else:
                    assert bys[offs[i]:offs[i + 1]].decode() == f"r{src}:{idx}"
                assert vval[i] == (0 if idx % 3 == 0 else 1)
        else:
            assert out.num_rows == 0

        # case 2: async murmur exchange, uniform scatter
        n2 = 200
        ids2 = np.arange(rank * 10_000, rank * 10_000 + n2, dtype=np.int64)
        b2 = Batch.from_dict({
            "id": ids2, "v": ids2.astype(np.float64),
            "k": np.zeros(n2, np.int32), "s": [f"x{i}" for i in ids2],
        }, schema)
        dest2 = torch.from_numpy((ids2 % world).astype(np.int64))
        h = exchange_batch_all_to_all_async(b2, dest2)
        # caller could decode the next unit here; then:
        out2 = h.wait()
        got2 = out2.columns["id"].data.numpy()
        assert (got2 % world == rank).all()
        offs2 = out2.columns["s"].offsets.numpy()
        bys2 = out2.columns["s"].bytes_.numpy().tobytes()
        for i, gid in enumerate(got2):
            assert bys2[offs2[i]:offs2[i + 1]].decode() == f"x{gid}"
        t2 = torch.tensor([out2.num_rows])
        dist.all_reduce(t2)
        assert int(t2.item()) == world * n2

        # case 3: list<float32> column (embedding tables through the
        # dataset exchange)
        schema3 = Schema([Field("id", "int64", False),
                          Field("emb", "list<float32>", True)])
        n3 = 48
        ids3 = np.arange(rank * 100, rank * 100 + n3, dtype=np.int64)
        b3 = Batch.from_dict({
            "id": ids3,
            "emb": [None if i % 7 == 0 else
                    [float(ids3[i]), float(i % 3)] * (i % 2 + 1)
                    for i in range(n3)],
        }, schema3)
        dest3 = torch.from_numpy((ids3 % world).astype(np.int64))
        out3 = exchange_batch_all_to_all(b3, dest3)
        gid = out3.columns["id"].data.numpy()
        assert (gid % world == rank).all()
        emb = out3.columns["emb"]
        offs3 = emb.offsets.numpy()
        vals3 = emb.data.numpy()
        assert int(offs3[-1]) == len(vals3)
        for i, g in enumerate(gid):
            idx_src = int(g % 100)
            if idx_src % 7 == 0:
                assert emb.validity is not None and emb.validity[i] == 0
            else:
                got_list = vals3[offs3[i]:offs3[i + 1]].tolist()
                expect = [float(g), float(idx_src % 3)] * (idx_src % 2 + 1)
                assert got_list == expect, (g, got_list, expect)

        # case 4: list<string> column (opaque prefixed-blob round)
        schema4 = Schema([Field("id", "int64", False),
                          Field("tags", "list<string>", True)])
        n4 = 40
        ids4 = np.arange(rank * 100, rank * 100 + n4, dtype=np.int64)
        b4 = Batch.from_dict({
            "id": ids4,
            "tags": [None if i % 6 == 0 else
                     [f"t{ids4[i]}", ""] [: i % 3] for i in range(n4)],
        }, schema4)
        dest4 = torch.from_numpy((ids4 % world).astype(np.int64))
        out4 = exchange_batch_all_to_all(b4, dest4)
        gid4 = out4.columns["id"].data.numpy()
        assert (gid4 % world == rank).all()
        tg = out4.columns["tags"]
        ro = tg.offsets.numpy()
        eo = tg.elem_offsets.numpy()
        bb = tg.bytes_.numpy().tobytes()
        for i, g in enumerate(gid4):
            idx_src = int(g % 100)
            if idx_src % 6 == 0:
                assert tg.validity is not None and tg.validity[i] == 0
            else:
                elems = [bb[eo[e]:eo[e + 1]].decode()
                         for e in range(ro[i], ro[i + 1])]
                assert elems == [f"t{g}", ""][: idx_src % 3], (g, elems)

        # case 5: struct + map columns (leaf-expanded exchange)
        schema5 = Schema([Field("id", "int64", False),
                          Field("st", "struct<a:int64,b:string>", True),
                          Field("mp", "map<string,int64>", True)])
        n5 = 30
        ids5 = np.arange(rank * 100, rank * 100 + n5, dtype=np.int64)
        st_rows = [None if i % 4 == 0 else
                   {"a": int(ids5[i]), "b": f"b{ids5[i]}"} for i in range(n5)]
        mp_rows = [None if i % 5 == 0 else
                   {f"k{j}": int(ids5[i]) + j for j in range(i % 3)}
                   for i in range(n5)]
        b5 = Batch.from_dict({"id": ids5, "st": st_rows, "mp": mp_rows},
                             schema5)
        dest5 = torch.from_numpy((ids5 % world).astype(np.int64))
        out5 = exchange_batch_all_to_all(b5, dest5)
        t5 = out5.to_arrow()
        for i, g in enumerate(t5.column("id").to_pylist()):
            idx_src = int(g % 100)
            st = t5.column("st").to_pylist()[i]
            mp_ = t5.column("mp").to_pylist()[i]
            if idx_src % 4 == 0:
                assert st is None
            else:
                assert st == {"a": g, "b": f"b{g}"}, (g, st)
            if idx_src % 5 == 0:
                assert mp_ is None
            else:
                assert mp_ == [(f"k{j}", g + j) for j in range(idx_src % 3)]
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()


def _spawn(fn, world, tmp_path):
    port = str(tmp_path / "rdv")
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        procs = [
            ctx.Process(target=fn, args=(r, world, port, str(tmp_path), results))
            for r in range(world)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
        for p in procs:
            assert p.exitcode == 0, f"worker failed (exit {p.exitcode})"
        for r in range(world):
            assert results.get(r) == "ok", f"rank {r} did not finish"


@pytest.mark.parametrize("fn", [_run_exchange, _run_table_shard, _run_rebalance])
def test_multiprocess_gloo(fn, tmp_path):
    _spawn(fn, 2, tmp_path)


@pytest.mark.parametrize("world", [4, 8])
def test_exchange_collectives_world_n(world, tmp_path):
    """gloo matrix for the exact bench collectives at world=4 and 8."""
    _spawn(_run_exchange_hard, world, tmp_path)


@pytest.mark.parametrize("world", [4])
def test_rebalance_world4(world, tmp_path):
    _spawn(_run_rebalance, world, tmp_path)
