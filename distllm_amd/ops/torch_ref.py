"""Plain-PyTorch fp32 reference implementations of every HIP op.

These are the ground truth for kernel numerics tests (GPU: HIP kernel vs
this in fp32) and the CPU execution path for the no-GPU test tier. They
define the op semantics exactly; the HIP kernels in csrc/ must match
within bf16 tolerance.
"""

from __future__ import annotations

import math

import torch
import torch.nn.functional as F


def norm_fwd(
    x: torch.Tensor,
    res: torch.Tensor | None,
    w: torch.Tensor,
    b: torch.Tensor | None,
    eps: float,
    rms: bool,
    return_sum: bool = False,
):
    xf = x.float()
    if res is not None:
        xf = xf + res.float()
    summed = xf
    if rms:
        inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
        out = xf * inv * w.float()
        if b is not None:
            out = out + b.float()
    else:
        mean = xf.mean(-1, keepdim=True)
        var = xf.var(-1, unbiased=False, keepdim=True)
        out = (xf - mean) * torch.rsqrt(var + eps) * w.float()
        if b is not None:
            out = out + b.float()
    out = out.to(x.dtype)
    if return_sum:
        return out, summed.to(x.dtype)
    return out


def bias_gelu(x: torch.Tensor, bias: torch.Tensor | None, erf: bool = False):
    xf = x.float()
    if bias is not None:
        xf = xf + bias.float()
    out = F.gelu(xf) if erf else F.gelu(xf, approximate="tanh")
    return out.to(x.dtype)


def swiglu(gate: torch.Tensor, up: torch.Tensor):
    return (F.silu(gate.float()) * up.float()).to(gate.dtype)


def bias_add(x: torch.Tensor, bias: torch.Tensor):
    return (x.float() + bias.float()).to(x.dtype)


def mean_pool(hidden: torch.Tensor, mask: torch.Tensor, normalize: bool):
    """Masked mean excluding BOS (pos 0) and the last real token (pos L-1).

    Role of reference distllm/embed/poolers/mean.py:13-49, with ONE
    deliberate divergence: the reference's `attention_mask[:, L-1] = 0`
    uses a VECTOR column index, so for a mixed-length batch it zeroes
    the column at every row's length-1 across ALL rows (excluding
    interior tokens of longer rows). We exclude each row's OWN last
    real token only — the per-row behavior the reference intends
    ("Set the attention mask to 0 for start and end tokens"). Equal
    results for uniform-length batches.
    hidden [R, S, H]; mask [R, S] int -> [R, H] fp32.
    """
    hf = hidden.float()
    m = mask.clone().to(hf.dtype)
    L = mask.sum(dim=1).long()
    m[:, 0] = 0
    rows = torch.arange(m.size(0), device=m.device)
    last = (L - 1).clamp(min=0)
    m[rows, last] = 0
    summed = (hf * m.unsqueeze(-1)).sum(dim=1)
    count = m.sum(dim=1, keepdim=True).clamp(min=1)
    out = summed / count
    if normalize:
        out = F.normalize(out, p=2, dim=-1)
    return out


def last_token_pool(
    hidden: torch.Tensor, mask: torch.Tensor, left_padded: bool, normalize: bool
):
    """reference distllm/embed/poolers/last_token.py:12-38 semantics."""
    hf = hidden.float()
    R, S, H = hf.shape
    if left_padded:
        out = hf[:, -1]
    else:
        idx = (mask.sum(dim=1).long() - 1).clamp(min=0)
        out = hf[torch.arange(R, device=hf.device), idx]
    if normalize:
        out = F.normalize(out, p=2, dim=-1)
    return out


def l2_normalize(x: torch.Tensor):
    return F.normalize(x.float(), p=2, dim=-1)


def adjacent_cosine(emb: torch.Tensor):
    """reference embed/embedders/semantic_chunk.py:24-55: 1 - cosine of
    adjacent rows."""
    e = emb.float()
    if e.size(0) < 2:
        return torch.empty(0, dtype=torch.float32, device=e.device)
    a, b = e[:-1], e[1:]
    num = (a * b).sum(-1)
    den = a.norm(dim=-1).clamp(min=1e-15) * b.norm(dim=-1).clamp(min=1e-15)
    return 1.0 - num / den


def gemm(A: torch.Tensor, B: torch.Tensor, bias=None, act: int = 0):
    """C = A @ B^T (+bias) (+activation). act: 0 none, 1 gelu_tanh,
    2 gelu_erf, 3 silu."""
    C = A.float() @ B.float().t()
    if bias is not None:
        C = C + bias.float()
    if act == 1:
        C = F.gelu(C, approximate="tanh")
    elif act == 2:
        C = F.gelu(C)
    elif act == 3:
        C = F.silu(C)
    return C.to(A.dtype)


def attn_fwd(
    Q: torch.Tensor,
    K: torch.Tensor,
    V: torch.Tensor,
    mask: torch.Tensor | None,
    causal: bool,
    scale: float,
):
    """BSHD layout; mask f32 [B, SKV] additive; GQA by head repeat."""
    B, SQ, H, D = Q.shape
    SKV, HKV = K.shape[1], K.shape[2]
    q = Q.float().permute(0, 2, 1, 3)  # B H S D
    k = K.float().permute(0, 2, 1, 3)
    v = V.float().permute(0, 2, 1, 3)
    if HKV != H:
        rep = H // HKV
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    s = (q @ k.transpose(-1, -2)) * scale  # B H SQ SKV
    if mask is not None:
        s = s + mask.float()[:, None, None, :]
    if causal:
        cm = torch.full((SQ, SKV), float("-inf"), device=s.device)
        cm = torch.triu(cm, diagonal=1)
        s = s + cm
    p = torch.softmax(s, dim=-1)
    p = torch.nan_to_num(p, nan=0.0)  # fully-masked rows -> 0
    o = p @ v
    return o.permute(0, 2, 1, 3).contiguous().to(Q.dtype)


def attn_varlen(Q, K, V, cu_seqlens, scale: float):
    """Packed [T, H(KV), D] varlen bidirectional attention reference:
    per-sequence slices through attn_fwd (fp32)."""
    cu = cu_seqlens.to("cpu", torch.int64).tolist()
    out = torch.empty_like(Q)
    for b in range(len(cu) - 1):
        lo, hi = cu[b], cu[b + 1]
        if hi <= lo:
            continue
        o = attn_fwd(Q[lo:hi][None], K[lo:hi][None], V[lo:hi][None],
                     None, False, scale)
        out[lo:hi] = o[0]
    return out


def rope_tables(max_pos: int, D: int, theta: float = 10000.0, device="cpu"):
    inv = 1.0 / (
        theta ** (torch.arange(0, D, 2, dtype=torch.float32, device=device) / D)
    )
    pos = torch.arange(max_pos, dtype=torch.float32, device=device)
    ang = pos[:, None] * inv[None, :]  # [S, D/2]
    return ang.cos().contiguous(), ang.sin().contiguous()


def rope_apply(x: torch.Tensor, cs: torch.Tensor, sn: torch.Tensor,
               pos: torch.Tensor, rope_heads: int = -1):
    """x [..., S, Hx, D] flattened rows [B,S,Hx,D]; pos int32 [B*S].
    Heads >= rope_heads (when >= 0) pass through unchanged."""
    shape = x.shape
    D = shape[-1]
    Hx = shape[-2]
    xf = x.float().reshape(-1, Hx, D)
    p = pos.long().reshape(-1)
    c = cs[p].unsqueeze(1)  # [rows, 1, D/2]
    s = sn[p].unsqueeze(1)
    x1, x2 = xf[..., : D // 2], xf[..., D // 2 :]
    out = torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)
    if 0 <= rope_heads < Hx:
        out[:, rope_heads:] = xf[:, rope_heads:]
    return out.reshape(shape).to(x.dtype)


def topk(scores: torch.Tensor, k: int):
    v, i = torch.topk(scores.float(), k, dim=-1)
    return v, i.int()


def hamming_score(qcodes: torch.Tensor, codes_t: torch.Tensor):
    """qcodes [Q, W] int64 (bit-packed), codes_t [W, N] int64 word-major."""
    Q, W = qcodes.shape
    N = codes_t.shape[1]
    out = torch.zeros(Q, N, dtype=torch.float32, device=qcodes.device)
    for w in range(W):
        x = qcodes[:, w : w + 1] ^ codes_t[w][None, :]
        # popcount on int64 via bit tricks (no torch.popcount on this ver)
        cnt = torch.zeros_like(x)
        for shift in range(0, 64, 8):
            byte = (x >> shift) & 0xFF
            cnt += _POPCNT_TABLE.to(x.device)[byte.long()]
        out -= cnt.float()
    return out


_POPCNT_TABLE = torch.tensor(
    [bin(i).count("1") for i in range(256)], dtype=torch.int64
)


def sample_tokens(logits: torch.Tensor, rand: torch.Tensor, temp: float,
                  min_p: float):
    lf = logits.float()
    if temp <= 0:
        return lf.argmax(dim=-1).int()
    mx = lf.max(dim=-1, keepdim=True).values
    if min_p > 0:
        cut = mx + temp * math.log(min_p)
        lf = torch.where(lf >= cut, lf, torch.full_like(lf, float("-inf")))
    p = torch.softmax((lf - mx) / temp, dim=-1)
    cdf = p.cumsum(dim=-1)
    r = rand.float().unsqueeze(-1)
    idx = (cdf < r).sum(dim=-1).clamp(max=lf.size(-1) - 1)
    return idx.int()


def paged_attn(q, kcache, vcache, block_table, ctx_len, scale):
    """q [B,H,D]; caches [nblk, HKV, BLK, D]; reference decode attention."""
    B, H, D = q.shape
    HKV = kcache.shape[1]
    BLK = kcache.shape[2]
    out = torch.empty_like(q)
    for b in range(B):
        L = int(ctx_len[b])
        nb = (L + BLK - 1) // BLK
        blocks = block_table[b, :nb].long()
        k = kcache[blocks].permute(1, 0, 2, 3).reshape(HKV, -1, D)[:, :L]
        v = vcache[blocks].permute(1, 0, 2, 3).reshape(HKV, -1, D)[:, :L]
        rep = H // HKV
        k = k.repeat_interleave(rep, dim=0).float()
        v = v.repeat_interleave(rep, dim=0).float()
        s = (q[b].float().unsqueeze(1) * k).sum(-1) * scale  # [H, L]
        p = torch.softmax(s, dim=-1)
        out[b] = (p.unsqueeze(-1) * v).sum(1).to(q.dtype)
    return out
