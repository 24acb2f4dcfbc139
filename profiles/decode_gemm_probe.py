"""Decode-shape GEMM probe: how close does hipBLASLt get to the
weight-streaming roofline at small M (batch) for Llama-3-8B layers?
Roofline: time >= weight_bytes / 6.3 TB/s.
"""

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))

import torch


def bench(fn, iters=50):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    M = int(sys.argv[1]) if len(sys.argv) > 1 else 128
    shapes = [
        ("qkv(fused)", 4096 + 2 * 1024, 4096),
        ("o", 4096, 4096),
        ("gate", 14336, 4096),
        ("down", 4096, 14336),
        ("lm_head", 128256, 4096),
    ]
    import distllm_amd.ops as ops

    for name, N, K in shapes:
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        t = bench(lambda: torch.nn.functional.linear(x, w))
        sk = ops._decode_splitk(M, N, K)
        td = bench(lambda: ops._hip_ops.decode_gemm(x, w, None, sk))
        wbytes = N * K * 2
        roof = wbytes / 6.3e12
        print(f"M{M} {name:<11} N{N} K{K}: blaslt {t * 1e6:7.1f}us "
              f"({100 * roof / t:4.1f}%) | decode sk{sk} {td * 1e6:7.1f}us "
              f"({100 * roof / td:4.1f}%)")


if __name__ == "__main__":
    main()
