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


def test_fuzz_varlen_attention(device):
    """Random dispersed length mixes (incl. zero-length and >128-block
    sequences) vs the per-sequence fp32 reference."""
    g = torch.Generator().manual_seed(321)
    for it in range(10):
        B = int(torch.randint(1, 7, (1,), generator=g))
        lens = [int(torch.randint(0, 400, (1,), generator=g))
                for _ in range(B)]
        if sum(lens) == 0:
            lens[0] = 5
        H = [2, 4, 8][int(torch.randint(0, 3, (1,), generator=g))]
        D = [64, 128][int(torch.randint(0, 2, (1,), generator=g))]
        gqa = int(torch.randint(0, 2, (1,), generator=g))
        HKV = H // 2 if gqa else H
        cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0)),
                          dtype=torch.int32, device=device)
        T = sum(lens)
        q = torch.randn(T, H, D, device=device, dtype=torch.bfloat16)
        k = torch.randn(T, HKV, D, device=device, dtype=torch.bfloat16)
        v = torch.randn_like(k)
        out = ops.attn_varlen(q, k, v, cu)
        rep = H // HKV
        ref = torch_ref.attn_varlen(
            q, k.repeat_interleave(rep, dim=1),
            v.repeat_interleave(rep, dim=1), cu, 1.0 / math.sqrt(D))
        _close(out, ref)


def test_fuzz_ivf_score(device):
    """Random chunk counts/caps/dims (x4-aligned) and probe lists with
    duplicates vs the einsum reference."""
    g = torch.Generator().manual_seed(654)
    for _ in range(10):
        nch = int(torch.randint(1, 12, (1,), generator=g))
        cap = int(torch.randint(1, 140, (1,), generator=g))
        H = 4 * int(torch.randint(1, 100, (1,), generator=g))
        Q = int(torch.randint(1, 9, (1,), generator=g))
        nprobe = int(torch.randint(1, min(nch, 4) + 1, (1,), generator=g))
        lens = torch.randint(0, cap + 1, (nch,), generator=g).int()
        slabs = torch.randn(nch, cap, H, generator=g)
        valid = torch.arange(cap)[None, :] < lens[:, None]
        slabs = (slabs * valid.unsqueeze(-1)).to(device)
        lens_d = lens.to(device)
        probes = torch.randint(0, nch, (Q, nprobe), generator=g) \
            .int().to(device)
        qv = torch.randn(Q, H, generator=g).to(device)
        got = ops.ivf_score(slabs, lens_d, probes, qv)
        ref = torch.einsum("qpcd,qd->qpc", slabs[probes.long()], qv)
        ref = ref.masked_fill(
            ~(torch.arange(cap, device=device)[None, None, :]
              < lens_d[probes.long()][:, :, None]), float("-inf"))
        fin = torch.isfinite(ref)
        assert torch.equal(torch.isfinite(got), fin)
        assert torch.allclose(got[fin], ref[fin], atol=2e-3, rtol=2e-3)
