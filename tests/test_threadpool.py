"""Concurrency stress for the native thread pool: concurrent read calls
from python threads (mirrors the GPU prefetch pipeline) must be safe."""

from concurrent.futures import ThreadPoolExecutor

import numpy as np

from lakesoul_amd.io.schema import Field, Schema
from lakesoul_amd.ops import cpp


def test_concurrent_read_unit_raw(catalog):
    t = catalog.create_table(
        "tp",
        Schema([Field("id", "int64", False), Field("v", "float64"), Field("s", "string")]),
        primary_keys=["id"],
        hash_bucket_num=4,
    )
    n = 20000
    rng = np.random.default_rng(0)
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": rng.normal(size=n), "s": [f"s{i}" for i in range(n)]})
    for it in range(4):
        ids = rng.choice(n, 2000, replace=False).astype(np.int64)
        t.upsert({"id": ids, "v": np.zeros(2000), "s": ["u"] * 2000})

    scan = t.scan(device="cpu")
    units = scan.plan()
    names = scan.read_cols

    def fetch(u):
        return cpp().read_unit_raw(u.files, names, 0, False)

    with ThreadPoolExecutor(max_workers=4) as ex:
        for _ in range(6):  # repeated rounds to shake out stragglers
            outs = list(ex.map(fetch, units * 2))
    total = sum(sum(o["file_rows"]) for o in outs) // 2
    assert total == n + 4 * 2000


def test_tsan_clean(tmp_path):
    """ThreadPool under ThreadSanitizer: no data races in the job-queue
    handoff (SURVEY.md §5.2 race-detection item)."""
    import subprocess, shutil, os

    src = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                       "csrc", "cpp", "tests", "threadpool_tsan.cc")
    exe = str(tmp_path / "tp_tsan")
    cc = shutil.which("g++")
    if cc is None:
        pytest.skip("no g++")
    b = subprocess.run([cc, "-O1", "-g", "-std=c++17", "-fsanitize=thread",
                        "-pthread", src, "-o", exe], capture_output=True, text=True)
    if b.returncode != 0:
        pytest.skip(f"tsan build unavailable: {b.stderr[-200:]}")
    env = dict(os.environ, TSAN_OPTIONS="halt_on_error=1", LAKESOUL_POOL_THREADS="8")
    r = subprocess.run([exe], capture_output=True, text=True, timeout=300, env=env)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "WARNING: ThreadSanitizer" not in r.stderr, r.stderr[:2000]
