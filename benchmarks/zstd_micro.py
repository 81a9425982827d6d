import numpy as np, torch, time, os, sys
sys.path.insert(0, "/root/repo")
from lakesoul_amd.ops import cpp, hip

rng = np.random.default_rng(0)
PAGE = 32768
kinds = {
  "i64_arange": np.arange(PAGE // 8, dtype=np.int64).tobytes(),
  "f64_normal": rng.normal(size=PAGE // 8).tobytes(),
  "i32_small": rng.integers(0, 500, PAGE // 4).astype(np.int32).tobytes(),
  "i64_rand1e12": rng.integers(0, 10**12, PAGE // 8).astype(np.int64).tobytes(),
}
for name, payload in kinds.items():
    comp = cpp().zstd_compress_ref(payload, 1)
    npages = 4096
    src = torch.from_numpy(np.frombuffer(comp * npages, dtype=np.uint8).copy()).cuda()
    dst = torch.zeros(PAGE * npages, dtype=torch.uint8, device="cuda")
    jobs = torch.tensor(
        [[i * len(comp), len(comp), i * PAGE, PAGE] for i in range(npages)],
        dtype=torch.int64, device="cuda")
    st = hip().zstd_decompress_into(src, jobs, dst); torch.cuda.synchronize()
    assert int((st != 0).sum()) == 0
    t0 = time.time()
    for _ in range(3):
        hip().zstd_decompress_into(src, jobs, dst)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / 3
    gbps = PAGE * npages / dt / 1e9
    print(f"{name}: ratio {len(payload)/len(comp):.2f}  {dt*1e3:7.2f} ms  {gbps:6.1f} GB/s  blocks={os.environ.get('LAKESOUL_ZSTD_BLOCKS','2048')}")
