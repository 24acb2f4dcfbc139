"""Bandwidth probe for the memory-bound HIP kernels: achieved TB/s vs
the ~6.3 TB/s achievable HBM ceiling (guide §2).

  python profiles/kernel_bw_probe.py
"""

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


def report(name, t, nbytes):
    print(f"{name:<34} {t * 1e6:9.1f} us   {nbytes / t / 1e12:5.2f} TB/s")


def main():
    assert torch.cuda.is_available() and ops.HAVE_HIP
    dev = "cuda"
    M, H = 262144, 768
    x = torch.randn(M, H, device=dev, dtype=torch.bfloat16)
    r = torch.randn_like(x)
    w = torch.randn(H, device=dev, dtype=torch.bfloat16)
    b = torch.randn(H, device=dev, dtype=torch.bfloat16)
    report("res+LayerNorm [262144,768]",
           bench(lambda: ops.layer_norm(x, w, b, res=r)), 3 * M * H * 2)

    x4 = torch.randn(65536, 4096, device=dev, dtype=torch.bfloat16)
    w4 = torch.randn(4096, device=dev, dtype=torch.bfloat16)
    report("res+RMSNorm [65536,4096]",
           bench(lambda: ops.rms_norm(x4, w4, res=x4)),
           3 * x4.numel() * 2)

    g = torch.randn(M, 3072, device=dev, dtype=torch.bfloat16)
    gb = torch.randn(3072, device=dev, dtype=torch.bfloat16)
    report("bias+GELU(erf) [262144,3072]",
           bench(lambda: ops.bias_gelu(g, gb, erf=True)),
           2 * g.numel() * 2)

    ga = torch.randn(65536, 14336, device=dev, dtype=torch.bfloat16)
    up = torch.randn_like(ga)
    report("SwiGLU [65536,14336]",
           bench(lambda: ops.swiglu(ga, up)), 3 * ga.numel() * 2)

    hid = torch.randn(512, 512, 768, device=dev, dtype=torch.bfloat16)
    mask = torch.ones(512, 512, device=dev, dtype=torch.int32)
    report("mean_pool+L2 [512,512,768]",
           bench(lambda: ops.mean_pool(hid, mask, normalize=True)),
           hid.numel() * 2)

    rope_x = torch.randn(64, 2048, 32, 128, device=dev,
                         dtype=torch.bfloat16)
    cs, sn = ops.rope_tables(4096, 128, device=dev)
    pos = torch.arange(2048, device=dev,
                       dtype=torch.int32).repeat(64)
    report("RoPE [64,2048,32,128]",
           bench(lambda: ops.rope_apply(rope_x, cs, sn, pos)),
           2 * rope_x.numel() * 2)

    sc = torch.randn(16, 1 << 21, device=dev)
    report("topk radix-select [16, 2M] k=32",
           bench(lambda: ops.topk(sc, 32)), 2 * sc.numel() * 4)

    qc = torch.randint(-(2**62), 2**62, (16, 12), device=dev)
    ct = torch.randint(-(2**62), 2**62, (12, 1 << 21), device=dev)
    report("hamming [16 x 2M x 768bit]",
           bench(lambda: ops.hamming_score(qc, ct)),
           16 * ct.numel() * 8)

    e = torch.randn(1 << 20, 768, device=dev)
    report("adjacent_cosine [1M,768]",
           bench(lambda: ops.adjacent_cosine(e)), 2 * e.numel() * 4)


if __name__ == "__main__":
    main()
