"""A/B LAKESOUL_FS_QB for the fastscan estimator (GPU box)."""
import os
import subprocess
import sys

for qb in ("0", "2", "4", "8", "4"):
    env = dict(os.environ, LAKESOUL_FS_QB=qb, LAKESOUL_TIMING="1")
    p = subprocess.run([sys.executable, "benchmarks/vector_bench.py",
                        "--n", "5000000", "--dim", "768"],
                       env=env, capture_output=True, text=True, timeout=1200)
    lines = p.stdout.splitlines()
    ests = [l for l in lines if "vb_fastscan_est" in l]
    qps = [l for l in lines if '"fastscan-rabitq4-hiC"' in l]
    import json
    q = json.loads(qps[0])["qps"] if qps else -1
    est = ests[0].split()[1] if ests else "?"
    print(f"QB={qb}: est={est} ms/4calls, hiC qps={q:.0f}", flush=True)
