"""Verify the fp32 reference ops (ops/torch_ref.py) against NAIVE
loop implementations at tiny sizes. torch_ref is the ground truth every
GPU numerics test compares against — a bug there would let a wrong
kernel 'pass', so the reference itself gets an independent anchor."""

import math

import torch

from distllm_amd.ops import torch_ref


def naive_attention(q, k, v, causal, scale):
    B, S, H, D = q.shape
    HKV = k.shape[2]
    rep = H // HKV
    out = torch.zeros_like(q, dtype=torch.float32)
    for b in range(B):
        for h in range(H):
            kh = h // rep
            for i in range(S):
                scores = []
                for j in range(S):
                    if causal and j > i:
                        scores.append(float("-inf"))
                    else:
                        scores.append(float(
                            q[b, i, h].double() @ k[b, j, kh].double())
                            * scale)
                p = torch.softmax(torch.tensor(scores), dim=0)
                acc = torch.zeros(D, dtype=torch.float64)
                for j in range(S):
                    if p[j] > 0:
                        acc += float(p[j]) * v[b, j, kh].double()
                out[b, i, h] = acc.float()
    return out


def test_ref_attention_vs_naive():
    torch.manual_seed(0)
    B, S, H, D, HKV = 1, 5, 2, 8, 1
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, HKV, D)
    v = torch.randn(B, S, HKV, D)
    scale = 1 / math.sqrt(D)
    for causal in (False, True):
        ref = torch_ref.attn_fwd(q, k, v, None, causal, scale)
        naive = naive_attention(q, k, v, causal, scale)
        assert torch.allclose(ref.float(), naive, atol=1e-5), causal


def test_ref_layernorm_vs_naive():
    torch.manual_seed(1)
    x = torch.randn(3, 16)
    res = torch.randn(3, 16)
    w = torch.randn(16)
    b = torch.randn(16)
    out = torch_ref.norm_fwd(x, res, w, b, 1e-5, rms=False)
    for i in range(3):
        row = (x[i] + res[i]).double()
        mu = row.mean()
        var = ((row - mu) ** 2).mean()
        want = ((row - mu) / (var + 1e-5).sqrt()) * w.double() + b.double()
        assert torch.allclose(out[i].double(), want, atol=1e-5)


def test_ref_rmsnorm_vs_naive():
    torch.manual_seed(2)
    x = torch.randn(2, 12)
    w = torch.randn(12)
    out = torch_ref.norm_fwd(x, None, w, None, 1e-5, rms=True)
    for i in range(2):
        row = x[i].double()
        inv = 1.0 / ((row ** 2).mean() + 1e-5).sqrt()
        want = row * inv * w.double()
        assert torch.allclose(out[i].double(), want, atol=1e-5)


def test_ref_paged_attn_vs_naive():
    torch.manual_seed(3)
    B, H, HKV, D, BLK, maxb = 2, 2, 1, 8, 4, 3
    kc = torch.randn(B * maxb + 1, HKV, BLK, D)
    vc = torch.randn_like(kc)
    bt = torch.arange(1, B * maxb + 1, dtype=torch.int32).reshape(B, maxb)
    ctx = torch.tensor([5, 11], dtype=torch.int32)
    q = torch.randn(B, H, D)
    scale = 1 / math.sqrt(D)
    out = torch_ref.paged_attn(q, kc, vc, bt, ctx, scale)
    for b in range(B):
        L = int(ctx[b])
        keys = torch.stack([
            kc[int(bt[b, t // BLK]), 0, t % BLK] for t in range(L)])
        vals = torch.stack([
            vc[int(bt[b, t // BLK]), 0, t % BLK] for t in range(L)])
        for h in range(H):
            s = (keys.double() @ q[b, h].double()) * scale
            p = torch.softmax(s, dim=0)
            want = (p[:, None] * vals.double()).sum(0)
            assert torch.allclose(out[b, h].double(), want, atol=1e-5)


def test_ref_mean_pool_vs_naive():
    torch.manual_seed(4)
    h = torch.randn(2, 6, 8)
    mask = torch.tensor([[1, 1, 1, 1, 0, 0], [1, 1, 1, 1, 1, 1]],
                        dtype=torch.int32)
    out = torch_ref.mean_pool(h, mask, normalize=False)
    # reference semantics: BOS and final attended token (EOS) excluded
    for b in range(2):
        L = int(mask[b].sum())
        inner = h[b, 1:L - 1].double()
        want = inner.mean(0)
        assert torch.allclose(out[b].double(), want, atol=1e-5)


def test_ref_topk_vs_sort():
    torch.manual_seed(5)
    scores = torch.randn(3, 50)
    v, i = torch_ref.topk(scores, 7)
    want_v, want_i = torch.sort(scores, dim=1, descending=True)
    assert torch.allclose(v, want_v[:, :7])
    got = scores.gather(1, i.long())
    assert torch.allclose(got, v)
