"""gemm16 (8-phase 256^2 template) vs gemm8 vs hipBLASLt on the
flagship encoder shapes (M=262144: PubMedBERT bs512 seq512).

Round-2 target: in-tree kernel wins >=80% of flagship
GEMM time at >=1173 TF effective (hipBLASLt's round-1 number).
"""

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))

import torch

import distllm_amd.ops as ops
from distllm_amd.ops import _hip_ops, torch_ref


def bench(fn, iters=20):
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
    ap.add_argument("--check", action="store_true")
    ap.add_argument("--only", type=str, default="",
                    help="run just gemm16 on ONE shape (qkv/ffn2/...) "
                         "a few iters — for rocprofv3 wrapping")
    args = ap.parse_args()
    if args.only:
        shapes = {"qkv": (2304, 768, 0), "attn_out": (768, 768, 0),
                  "ffn1": (3072, 768, 2), "ffn2": (768, 3072, 0)}
        n, k, act = shapes[args.only]
        A = torch.randn(args.m, k, device="cuda",
                        dtype=torch.bfloat16) * 0.3
        B = torch.randn(n, k, device="cuda", dtype=torch.bfloat16) * 0.3
        from distllm_amd.ops import _hip_ops as ho
        for _ in range(5):
            ho.gemm16_bf16(A, B, None, act)
        torch.cuda.synchronize()
        return
    assert torch.cuda.is_available() and ops.HAVE_HIP
    M = args.m
    shapes = [
        ("qkv", M, 2304, 768, 0),
        ("attn_out", M, 768, 768, 0),
        ("ffn1+gelu", M, 3072, 768, ops.ACT_GELU_ERF),
        ("ffn2", M, 768, 3072, 0),
    ]
    print(f"{'shape':>10} {'blaslt':>8} {'blt_fus':>8} {'gemm8':>8} "
          f"{'gemm16':>8}   (TF/s; * = winner)")
    for name, m, n, k, act in shapes:
        A = torch.randn(m, k, device="cuda", dtype=torch.bfloat16) * 0.3
        B = torch.randn(n, k, device="cuda", dtype=torch.bfloat16) * 0.3
        bias = torch.randn(n, device="cuda", dtype=torch.bfloat16)
        tf = 2.0 * m * n * k / 1e12

        if args.check:
            # 32768 rows -> nwg > 256 so persistent multi-tile runs and
            # the cross-tile prefetch path are exercised
            got = _hip_ops.gemm16_bf16(A[:32768], B, bias, act)
            ref = torch_ref.gemm(A[:32768], B, bias, act=act)
            err = (got.float() - ref.float()).abs()
            tol = 5e-2 + 5e-2 * ref.float().abs()
            bad = (err > tol).float().mean().item()
            print(f"  [check {name}] frac_bad={bad:.2e} "
                  f"max_err={err.max().item():.3f}")

        def blaslt():
            out = torch.nn.functional.linear(A, B, bias)
            if act:
                out = ops.bias_gelu(out, None, erf=True)
            return out

        times = {
            "blaslt": bench(blaslt),
            "blt_fus": bench(lambda: ops._blt_fused_path(A, B, bias, act)),
            "gemm8": bench(lambda: _hip_ops.gemm8_bf16(A, B, bias, act)),
            "gemm16": bench(lambda: _hip_ops.gemm16_bf16(A, B, bias, act)),
        }
        best = min(times, key=times.get)
        row = " ".join(
            f"{tf / t:7.0f}{'*' if kk == best else ' '}"
            for kk, t in times.items()
        )
        print(f"{name:>10} {row}")


if __name__ == "__main__":
    main()
