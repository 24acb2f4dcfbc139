"""GPU edge-case robustness: odd sequence lengths, tiny batches,
k > N, extreme GQA ratios — shapes the block/tile geometry must guard."""

import math

import pytest
import torch

import distllm_amd.ops as ops
from distllm_amd.ops import torch_ref

pytestmark = pytest.mark.gpu


def _close(a, b, atol=3e-2, rtol=3e-2):
    err = (a.float().cpu() - b.float().cpu()).abs()
    tol = atol + rtol * b.float().cpu().abs()
    assert (err > tol).float().mean().item() < 1e-3


@pytest.mark.parametrize("S", [1, 7, 33, 63, 65, 129, 513])
def test_attention_odd_lengths(device, S):
    torch.manual_seed(S)
    q = torch.randn(1, S, 2, 64, device=device, dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    out = ops.attn_fwd(q, k, v, causal=True)
    ref = torch_ref.attn_fwd(q, k, v, None, True, 1.0 / math.sqrt(64))
    _close(out, ref)


def test_attention_extreme_gqa(device):
    torch.manual_seed(1)
    # 32 query heads on 1 kv head
    q = torch.randn(2, 96, 32, 128, device=device, dtype=torch.bfloat16)
    k = torch.randn(2, 96, 1, 128, device=device, dtype=torch.bfloat16)
    v = torch.randn_like(k)
    out = ops.attn_fwd(q, k, v, causal=False)
    ref = torch_ref.attn_fwd(q, k, v, None, False, 1.0 / math.sqrt(128))
    _close(out, ref)


def test_attention_single_row_batch(device):
    torch.manual_seed(2)
    q = torch.randn(1, 1, 12, 64, device=device, dtype=torch.bfloat16)
    kv = torch.randn(1, 40, 12, 64, device=device, dtype=torch.bfloat16)
    out = ops.attn_fwd(q, kv, kv.clone())
    ref = torch_ref.attn_fwd(q, kv, kv.clone(), None, False,
                             1.0 / math.sqrt(64))
    _close(out, ref)


def test_topk_k_larger_than_n(device):
    scores = torch.randn(3, 10, device=device)
    v, i = ops.topk(scores, 32)
    rv, _ = torch_ref.topk(scores, 10)
    assert torch.allclose(v[:, :10].cpu(), rv.cpu(), atol=1e-6)
    assert (v[:, 10:] == float("-inf")).all()
    assert (i[:, 10:] == -1).all()


def test_topk_tiny_and_ties(device):
    scores = torch.zeros(2, 1000, device=device)  # fully degenerate
    v, i = ops.topk(scores, 8)
    assert (v == 0).all()
    assert (i >= 0).all() and (i < 1000).all()


def test_norm_odd_rows(device):
    x = torch.randn(1, 768, device=device, dtype=torch.bfloat16)
    w = torch.randn(768, device=device, dtype=torch.bfloat16)
    out = ops.rms_norm(x, w)
    ref = torch_ref.norm_fwd(x, None, w, None, 1e-5, rms=True)
    _close(out, ref)


def test_pool_single_token_rows(device):
    # rows where L=1 or L=2: the BOS/EOS exclusion must clamp count >= 1
    h = torch.randn(4, 8, 768, device=device, dtype=torch.bfloat16)
    mask = torch.zeros(4, 8, dtype=torch.int32, device=device)
    mask[0, :1] = 1
    mask[1, :2] = 1
    mask[2, :3] = 1
    mask[3, :8] = 1
    out = ops.mean_pool(h, mask, normalize=False)
    ref = torch_ref.mean_pool(h, mask, normalize=False)
    _close(out, ref, atol=5e-3, rtol=5e-3)
    assert torch.isfinite(out).all()


def test_paged_attention_short_contexts(device):
    torch.manual_seed(3)
    B, H, HKV, D, BLK, maxb = 4, 4, 2, 128, 16, 2
    kc = torch.randn(B * maxb, HKV, BLK, D, device=device,
                     dtype=torch.bfloat16)
    vc = torch.randn_like(kc)
    bt = torch.arange(B * maxb, device=device,
                      dtype=torch.int32).reshape(B, maxb)
    ctx = torch.tensor([1, 2, 17, 31], device=device, dtype=torch.int32)
    q = torch.randn(B, H, D, device=device, dtype=torch.bfloat16)
    out = ops.paged_attn(q, kc, vc, bt, ctx)
    ref = torch_ref.paged_attn(q, kc, vc, bt, ctx, 1.0 / math.sqrt(D))
    _close(out, ref)


def test_paged_attention_long_context_small_batch(device):
    """B=1 long context exercises the flash-decoding split path
    (context chunks + combine pass)."""
    torch.manual_seed(5)
    B, H, HKV, D, BLK = 1, 32, 8, 128, 16
    ctx_n = 1999
    maxb = (ctx_n + BLK - 1) // BLK + 1
    kc = torch.randn(B * maxb + 1, HKV, BLK, D, device=device,
                     dtype=torch.bfloat16)
    vc = torch.randn_like(kc)
    bt = torch.arange(1, B * maxb + 1, device=device,
                      dtype=torch.int32).reshape(B, maxb)
    ctx = torch.tensor([ctx_n], device=device, dtype=torch.int32)
    q = torch.randn(B, H, D, device=device, dtype=torch.bfloat16)
    out = ops.paged_attn(q, kc, vc, bt, ctx)
    ref = torch_ref.paged_attn(q, kc, vc, bt, ctx, 1.0 / math.sqrt(D))
    _close(out, ref)
