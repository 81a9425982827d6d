"""A/B hybrid host/GPU zstd at 32KB pages on the headline bench."""
import json
import os
import subprocess
import sys

for pages, frac in ((32768, "0.0"), (32768, "0.2"), (32768, "0.35"),
                    (32768, "0.5"), (131072, "0.0")):
    env = dict(os.environ, LAKESOUL_PAGE_BYTES=str(pages),
               LAKESOUL_GPU_ZSTD_FRAC=frac)
    p = subprocess.run([sys.executable, "bench.py", "--steps", "10",
                        "--warmup", "3"], env=env, capture_output=True,
                       text=True, timeout=400)
    line = [l for l in p.stdout.strip().splitlines() if l.startswith("{")]
    if not line:
        print(f"pages={pages} frac={frac}: FAILED\n{p.stderr[-400:]}")
        continue
    d = json.loads(line[-1])
    print(f"pages={pages} frac={frac}: {d['ms_per_step']:.1f} ms/step, "
          f"{d['value']/1e6:.1f}M rows/s", flush=True)
