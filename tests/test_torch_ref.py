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


def test_ref_gelu_vs_closed_form():
    x = torch.linspace(-4, 4, 33)
    out_erf = torch_ref.bias_gelu(x, None, erf=True)
    want = x * 0.5 * (1 + torch.erf(x / math.sqrt(2)))
    assert torch.allclose(out_erf, want, atol=1e-6)
    out_tanh = torch_ref.bias_gelu(x, None, erf=False)
    inner = math.sqrt(2 / math.pi) * (x + 0.044715 * x ** 3)
    want_t = 0.5 * x * (1 + torch.tanh(inner))
    assert torch.allclose(out_tanh, want_t, atol=1e-5)
    # swiglu = silu(gate) * up
    g, u = torch.randn(4, 8), torch.randn(4, 8)
    assert torch.allclose(torch_ref.swiglu(g, u),
                          torch.nn.functional.silu(g) * u, atol=1e-6)


def test_ref_rope_rotation_properties():
    """RoPE must preserve norms and make QK^T depend only on relative
    position."""
    torch.manual_seed(6)
    D = 16
    cs, sn = torch_ref.rope_tables(64, D)
    x = torch.randn(1, 1, 1, D)
    for p in (0, 5, 31):
        r = torch_ref.rope_apply(x, cs, sn, torch.tensor([p]))
        assert torch.allclose(r.norm(), x.norm(), atol=1e-5)
    q = torch.randn(1, 1, 1, D)
    k = torch.randn(1, 1, 1, D)

    def dot(pq, pk):
        rq = torch_ref.rope_apply(q, cs, sn, torch.tensor([pq]))
        rk = torch_ref.rope_apply(k, cs, sn, torch.tensor([pk]))
        return float(rq.flatten() @ rk.flatten())

    assert abs(dot(3, 1) - dot(10, 8)) < 1e-4   # same offset
    assert abs(dot(3, 1) - dot(4, 1)) > 1e-6    # different offset


def test_ref_hamming_vs_python_popcount():
    torch.manual_seed(7)
    Q, W, N = 3, 2, 5
    qc = torch.randint(-(2 ** 62), 2 ** 62, (Q, W), dtype=torch.int64)
    ct = torch.randint(-(2 ** 62), 2 ** 62, (W, N), dtype=torch.int64)
    out = torch_ref.hamming_score(qc, ct)
    for qi in range(Q):
        for n in range(N):
            ham = sum(
                bin((int(qc[qi, w]) ^ int(ct[w, n])) & (2 ** 64 - 1))
                .count("1") for w in range(W))
            assert out[qi, n] == -ham


def test_ref_sampling_distribution():
    """Temperature sampling via inverse-CDF must hit each token with
    ~softmax probability; min_p masks low-probability tokens."""
    torch.manual_seed(8)
    logits = torch.tensor([[2.0, 1.0, -1.0, -10.0]]).repeat(4000, 1)
    rand = torch.rand(4000)
    ids = torch_ref.sample_tokens(logits, rand, temp=1.0, min_p=0.0)
    freq = torch.bincount(ids.long(), minlength=4).float() / 4000
    want = torch.softmax(logits[0], -1)
    assert (freq - want).abs().max() < 0.03
    # min_p=0.2: tokens below 0.2*max_prob are never sampled
    ids2 = torch_ref.sample_tokens(logits, rand, temp=1.0, min_p=0.2)
    assert set(ids2.unique().tolist()) <= {0, 1}
    # temp=0 is argmax
    ids3 = torch_ref.sample_tokens(logits, rand, temp=0.0, min_p=0.5)
    assert (ids3 == 0).all()


def test_ref_l2_and_adjacent_cosine():
    torch.manual_seed(9)
    x = torch.randn(5, 8)
    n = torch_ref.l2_normalize(x)
    assert torch.allclose(n.norm(dim=1), torch.ones(5), atol=1e-5)
    d = torch_ref.adjacent_cosine(n)
    for i in range(4):
        cos = float(n[i] @ n[i + 1])
        assert abs(float(d[i]) - (1 - cos)) < 1e-5


def test_ref_last_token_pool_vs_naive():
    torch.manual_seed(10)
    h = torch.randn(2, 6, 8)
    mask = torch.tensor([[1, 1, 1, 0, 0, 0], [1, 1, 1, 1, 1, 1]],
                        dtype=torch.int32)
    out = torch_ref.last_token_pool(h, mask, left_padded=False,
                                    normalize=False)
    assert torch.allclose(out[0], h[0, 2])
    assert torch.allclose(out[1], h[1, 5])
    # left padding: last token is always the final position
    mask_l = torch.tensor([[0, 0, 0, 1, 1, 1]], dtype=torch.int32)
    out_l = torch_ref.last_token_pool(h[:1], mask_l, left_padded=True,
                                      normalize=False)
    assert torch.allclose(out_l[0], h[0, 5])


def test_attn_varlen_ref_matches_padded():
    """torch_ref.attn_varlen == padded attn_fwd with key-pad mask on
    the valid positions."""
    import math

    from distllm_amd.ops import torch_ref, varlen_block_map

    torch.manual_seed(40)
    lens = [7, 20, 1, 13]
    B, S, H, D = len(lens), max(lens), 3, 16
    q = torch.randn(B, S, H, D, dtype=torch.bfloat16)
    k = torch.randn(B, S, H, D, dtype=torch.bfloat16)
    v = torch.randn(B, S, H, D, dtype=torch.bfloat16)
    maskb = torch.zeros(B, S, dtype=torch.bool)
    for r, L in enumerate(lens):
        maskb[r, :L] = True
    mask_f = torch.where(maskb, 0.0, float("-inf")).float()
    scale = 1.0 / math.sqrt(D)
    padded = torch_ref.attn_fwd(q, k, v, mask_f, False, scale)

    cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0)),
                      dtype=torch.int32)
    packed = torch_ref.attn_varlen(q[maskb], k[maskb], v[maskb], cu, scale)
    assert torch.allclose(packed.float(), padded[maskb].float(),
                          atol=2e-2, rtol=2e-2)

    bs, bq = varlen_block_map(cu)
    assert bs.tolist() == [0, 1, 2, 3] and bq.tolist() == [0, 0, 0, 0]
    bs2, bq2 = varlen_block_map(
        torch.tensor([0, 130, 386], dtype=torch.int32))
    assert bs2.tolist() == [0, 0, 1, 1] and bq2.tolist() == [0, 1, 0, 1]


def test_varlen_block_map_edge_cases():
    """Zero sequences, zero-length sequences, and exact block
    multiples all yield a consistent (seq, qblock) map."""
    from distllm_amd.ops import varlen_block_map

    def m(cu):
        bs, bq = varlen_block_map(
            torch.tensor(cu, dtype=torch.int32), 64)
        return bs.tolist(), bq.tolist()

    assert m([0]) == ([], [])
    assert m([0, 1]) == ([0], [0])
    assert m([0, 64]) == ([0], [0])          # exact multiple: 1 block
    assert m([0, 65]) == ([0, 0], [0, 1])
    # zero-length sequence in the middle is skipped entirely
    assert m([0, 64, 64, 200]) == ([0, 2, 2, 2], [0, 0, 1, 2])


def test_quantize_fp8_rows_cpu():
    """Per-channel fp8-e4m3 weight quantization: dequant error bounded
    by the e4m3 step (~6% relative at per-row absmax scaling), exact
    zeros, and scale shape/positivity."""
    from distllm_amd.ops import quantize_fp8_rows

    torch.manual_seed(80)
    w = torch.randn(16, 64, dtype=torch.bfloat16) * 3.0
    w[0] = 0.0
    wq, sc = quantize_fp8_rows(w)
    assert wq.dtype == torch.uint8 and wq.shape == w.shape
    assert sc.dtype == torch.float32 and sc.shape == (16,)
    assert (sc > 0).all()
    wd = wq.view(torch.float8_e4m3fn).float() * sc[:, None]
    assert torch.equal(wd[0], torch.zeros(64))
    rel = (wd[1:] - w[1:].float()).abs().max(1).values / \
        w[1:].float().abs().max(1).values
    assert rel.max().item() < 0.08
