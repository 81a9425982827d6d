"""A/B prefetch depth / streams for the headline bench (run on GPU box)."""
import json
import os
import subprocess
import sys

for depth, streams in ((8, 2), (10, 2), (12, 2), (16, 2), (8, 2)):
    env = dict(os.environ, LAKESOUL_SCAN_DEPTH=str(depth),
               LAKESOUL_SCAN_STREAMS=str(streams))
    p = subprocess.run([sys.executable, "bench.py", "--steps", "10",
                        "--warmup", "3"], env=env, capture_output=True,
                       text=True, timeout=400)
    line = [l for l in p.stdout.strip().splitlines() if l.startswith("{")]
    if not line:
        print(f"depth={depth} streams={streams}: FAILED\n{p.stderr[-500:]}")
        continue
    d = json.loads(line[-1])
    print(f"depth={depth} streams={streams}: {d['ms_per_step']:.1f} ms/step, "
          f"{d['value']/1e6:.1f}M rows/s", flush=True)
