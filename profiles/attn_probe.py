"""Attention kernel probe: times ops.attn_fwd on the flagship embed
shape and prints effective TFLOP/s. Used with rocprofv3 for PMC runs.

  python profiles/attn_probe.py [--iters 50] [--d 64] [--causal]
"""

import argparse
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).parent.parent))
import time

import torch

import distllm_amd.ops as ops


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--b", type=int, default=512)
    ap.add_argument("--s", type=int, default=512)
    ap.add_argument("--h", type=int, default=12)
    ap.add_argument("--hkv", type=int, default=0)
    ap.add_argument("--d", type=int, default=64)
    ap.add_argument("--causal", action="store_true")
    ap.add_argument("--nomask", action="store_true")
    args = ap.parse_args()

    assert torch.cuda.is_available() and ops.HAVE_HIP
    B, S, H, D = args.b, args.s, args.h, args.d
    HKV = args.hkv or H
    torch.manual_seed(0)
    q = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, S, HKV, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, S, HKV, D, device="cuda", dtype=torch.bfloat16)
    mask = None
    if not args.nomask:
        mask = torch.zeros(B, S, device="cuda", dtype=torch.float32)

    for _ in range(5):
        o = ops.attn_fwd(q, k, v, mask=mask, causal=args.causal)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        o = ops.attn_fwd(q, k, v, mask=mask, causal=args.causal)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.iters
    flops = 4.0 * B * H * S * S * D  # QK + PV, 2 flops/MAC each
    if args.causal:
        flops /= 2
    print(f"attn B{B} S{S} H{H}/{HKV} D{D} causal={args.causal}: "
          f"{dt * 1e3:.3f} ms  {flops / dt / 1e12:.1f} TF  "
          f"sum={o.float().abs().mean().item():.4f}")


if __name__ == "__main__":
    main()
