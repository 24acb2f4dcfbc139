"""Per-shape bandwidth probe of the W8A16 gemv_fp8 kernel on the
llama3-8b projection shapes (B=1 decode regime)."""

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))

import torch

from distllm_amd import ops  # noqa: E402

for N, K in [(6144, 4096), (4096, 4096), (28672, 4096), (4096, 14336)]:
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    wq, sc = ops.quantize_fp8_rows(w)
    x = torch.randn(1, K, device="cuda", dtype=torch.bfloat16)
    for _ in range(10):
        ops.gemv_fp8(x, wq, sc)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(200):
        ops.gemv_fp8(x, wq, sc)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 200
    print(f"N={N} K={K}: {dt*1e6:6.1f} us  {N*K/dt/1e12:5.2f} TB/s eff")
