import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from sparkrdma_amd.ops import load

m = load()
NR = 256
region_elems = 16 << 20          # 128 MiB per region x 256 = 32 GiB arena? too big
region_elems = 4 << 20           # 32 MiB x 256 = 8 GiB
out = torch.empty(NR * region_elems, dtype=torch.int64, device="cuda")
s = torch.cuda.current_stream().cuda_stream
total_bytes = 2 << 30            # 2 GiB written per trial
for burst_bytes in (64, 128, 256, 512, 1024, 4096, 65536):
    be = burst_bytes // 8
    grid = 4096
    bpb = max(1, total_bytes // (burst_bytes * grid))
    # warmup
    m.probe_scatter_write(out.data_ptr(), region_elems, NR, be, bpb, grid, s)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 5
    for _ in range(iters):
        m.probe_scatter_write(out.data_ptr(), region_elems, NR, be, bpb, grid, s)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    gb = grid * bpb * burst_bytes / 1e9
    print(f"burst {burst_bytes:6d} B: {gb/dt:7.1f} GB/s  ({gb:.2f} GB in {dt*1e3:.2f} ms)")
