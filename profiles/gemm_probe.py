"""GEMM probe: hand-written fused MFMA GEMM (gemm.hip) vs hipBLASLt
(torch F.linear [+ separate fused-gelu kernel]) on the flagship encoder
layer shapes. Decides the DISTLLM_GEMM dispatch default.
"""

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))

import torch

import distllm_amd.ops as ops


def bench(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--m", type=int, default=262144)
    args = ap.parse_args()
    assert torch.cuda.is_available() and ops.HAVE_HIP
    M = args.m
    shapes = [
        ("qkv", M, 2304, 768, 0),
        ("attn_out", M, 768, 768, 0),
        ("ffn1+gelu", M, 3072, 768, ops.ACT_GELU_ERF),
        ("ffn2", M, 768, 3072, 0),
    ]
    for name, m, n, k, act in shapes:
        A = torch.randn(m, k, device="cuda", dtype=torch.bfloat16) * 0.3
        B = torch.randn(n, k, device="cuda", dtype=torch.bfloat16) * 0.3
        bias = torch.randn(n, device="cuda", dtype=torch.bfloat16)
        tf = 2.0 * m * n * k / 1e12

        def blaslt():
            out = torch.nn.functional.linear(A, B, bias)
            if act:
                out = ops.bias_gelu(out, None, erf=True)
            return out

        def hip():
            return ops.gemm(A, B, bias, act=act)

        def hip8():
            return ops._hip_ops.gemm8_bf16(A, B, bias, act)

        t1 = bench(blaslt)
        t2 = bench(hip)
        t3 = bench(hip8)
        r8 = hip8()[:256]
        ref8 = ops.torch_ref.gemm(A[:256].float().bfloat16(), B, bias, act)
        err8 = (r8.float() - ref8.float()).abs().max().item()
        r1 = ops.torch_ref.gemm(A[:256].float().bfloat16(), B, bias, act)
        r2 = hip()[:256]
        err = (r2.float() - r1.float()).abs().max().item()
        print(f"{name:<10} M{m} N{n} K{k}: blaslt {t1 * 1e3:7.3f} ms "
              f"({tf / t1:6.0f} TF) | g1 {t2 * 1e3:7.3f} ({tf / t2:5.0f} TF) "
              f"| g8 {t3 * 1e3:7.3f} ({tf / t3:5.0f} TF) "
              f"| err {err:.3f}/{err8:.3f}")


if __name__ == "__main__":
    main()
