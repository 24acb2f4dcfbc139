"""Seeded random-shape fuzz of the core HIP kernels against the fp32
torch reference: many odd shapes per op, one test per op family.
Complements the fixed-shape numerics tests (test_ops_gpu.py) with
coverage of the block/tile-geometry edge space."""

import math

import pytest
import torch

import distllm_amd.ops as ops
from distllm_amd.ops import torch_ref

pytestmark = pytest.mark.gpu


def _close(a, b, atol=3e-2, rtol=3e-2, frac=1e-3):
    err = (a.float().cpu() - b.float().cpu()).abs()
    tol = atol + rtol * b.float().cpu().abs()
    assert (err > tol).float().mean().item() < frac, \
        f"max err {err.max().item():.4f}"


def test_fuzz_attention(device):
    g = torch.Generator().manual_seed(123)
    for _ in range(12):
        B = int(torch.randint(1, 4, (1,), generator=g))
        S = int(torch.randint(1, 300, (1,), generator=g))
        H = int(torch.randint(1, 9, (1,), generator=g))
        D = [32, 64, 96, 128][int(torch.randint(0, 4, (1,), generator=g))]
        causal = bool(torch.randint(0, 2, (1,), generator=g))
        gqa = int(torch.randint(0, 2, (1,), generator=g))
        HKV = max(1, H // 2) if gqa and H % 2 == 0 else H
        q = torch.randn(B, S, H, D, device=device, dtype=torch.bfloat16)
        k = torch.randn(B, S, HKV, D, device=device, dtype=torch.bfloat16)
        v = torch.randn_like(k)
        out = ops.attn_fwd(q, k, v, causal=causal)
        ref = torch_ref.attn_fwd(q, k, v, None, causal, 1 / math.sqrt(D))
        _close(out, ref)


def test_fuzz_norms(device):
    g = torch.Generator().manual_seed(7)
    for _ in range(16):
        R = int(torch.randint(1, 5000, (1,), generator=g))
        H = 8 * int(torch.randint(1, 1025, (1,), generator=g))
        rms = bool(torch.randint(0, 2, (1,), generator=g))
        res = bool(torch.randint(0, 2, (1,), generator=g))
        x = torch.randn(R, H, device=device, dtype=torch.bfloat16)
        r = torch.randn_like(x) if res else None
        w = torch.randn(H, device=device, dtype=torch.bfloat16)
        b = None if rms else torch.randn_like(w)
        out = ops.norm_fwd(x, res=r, w=w, b=b, rms=rms)
        ref = torch_ref.norm_fwd(x, r, w, b, 1e-5, rms)
        _close(out, ref)


def test_fuzz_linear(device):
    g = torch.Generator().manual_seed(17)
    for _ in range(10):
        M = 2 ** int(torch.randint(0, 12, (1,), generator=g))
        N = 64 * int(torch.randint(1, 33, (1,), generator=g))
        K = 64 * int(torch.randint(1, 25, (1,), generator=g))
        act = [ops.ACT_NONE, ops.ACT_GELU_ERF,
               ops.ACT_GELU_TANH][int(torch.randint(0, 3, (1,),
                                                    generator=g))]
        bias = bool(torch.randint(0, 2, (1,), generator=g))
        x = torch.randn(M, K, device=device, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=device, dtype=torch.bfloat16) * 0.05
        b = torch.randn(N, device=device, dtype=torch.bfloat16) \
            if bias else None
        out = ops.linear(x, w, b, act=act)
        ref = torch_ref.gemm(x, w, b, act=act)
        _close(out, ref, atol=5e-2, rtol=5e-2)


def test_fuzz_topk(device):
    g = torch.Generator().manual_seed(29)
    for _ in range(10):
        Q = int(torch.randint(1, 20, (1,), generator=g))
        N = int(torch.randint(2, 200_000, (1,), generator=g))
        k = int(torch.randint(1, 65, (1,), generator=g))
        scores = torch.randn(Q, N, device=device)
        v, i = ops.topk(scores, k)
        rv, _ = torch_ref.topk(scores, min(k, N))
        kk = min(k, N)
        assert torch.allclose(v[:, :kk].cpu(), rv.cpu(), atol=1e-5), \
            (Q, N, k)
        got = scores.gather(1, i[:, :kk].long().clamp(min=0))
        assert torch.allclose(got.cpu(), v[:, :kk].cpu(), atol=1e-5)


def test_fuzz_paged_attention(device):
    g = torch.Generator().manual_seed(31)
    for _ in range(8):
        B = int(torch.randint(1, 9, (1,), generator=g))
        H = [1, 2, 4, 8][int(torch.randint(0, 4, (1,), generator=g))]
        HKV = H if bool(torch.randint(0, 2, (1,), generator=g)) \
            else max(1, H // 2)
        D = [64, 128][int(torch.randint(0, 2, (1,), generator=g))]
        BLK = 16
        maxb = int(torch.randint(1, 9, (1,), generator=g))
        kc = torch.randn(B * maxb + 1, HKV, BLK, D, device=device,
                         dtype=torch.bfloat16)
        vc = torch.randn_like(kc)
        bt = torch.arange(1, B * maxb + 1, device=device,
                          dtype=torch.int32).reshape(B, maxb)
        ctx = torch.randint(1, maxb * BLK + 1, (B,), generator=g)\
            .to(device=device, dtype=torch.int32)
        q = torch.randn(B, H, D, device=device, dtype=torch.bfloat16)
        out = ops.paged_attn(q, kc, vc, bt, ctx)
        ref = torch_ref.paged_attn(q, kc, vc, bt, ctx, 1 / math.sqrt(D))
        _close(out, ref)
