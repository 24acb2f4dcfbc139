"""ESM protein-language-model encoders on the distllm-amd HIP ops.

Covers the reference's protein path (reference:
distllm/embed/encoders/esm2.py, esmc.py — which delegated to HF
EsmForMaskedLM / faesm flash-attn / EvolutionaryScale ESMC):

  - Esm2Encoder: pre-LN transformer, rotary q/k, erf-GELU FFN,
    token-dropout embedding rescale, final LayerNorm (HF EsmModel
    forward semantics).
  - EsmCEncoder: pre-LN transformer, rotary q/k, SwiGLU FFN (ESMC
    "Cambrian" family; dims 960/30/15 for 300m, 1152/36/18 for 600m).

bf16 weights, fp32 accumulation in kernels.
"""

from __future__ import annotations

import math
import os

import torch
from torch import nn

from .. import ops
from ..ops import torch_ref
from .config import ModelDims

ESM_MASK_TOKEN = 32  # <mask> id in the standard ESM2 alphabet


class Esm2Layer(nn.Module):
    def __init__(self, d: ModelDims):
        super().__init__()
        H = d.hidden_size
        self.d = d
        self.ln1_w = nn.Parameter(torch.empty(H))
        self.ln1_b = nn.Parameter(torch.empty(H))
        self.wqkv = nn.Parameter(torch.empty(3 * H, H))
        self.bqkv = nn.Parameter(torch.empty(3 * H))
        self.wo = nn.Parameter(torch.empty(H, H))
        self.bo = nn.Parameter(torch.empty(H))
        self.ln2_w = nn.Parameter(torch.empty(H))
        self.ln2_b = nn.Parameter(torch.empty(H))
        self.w1 = nn.Parameter(torch.empty(d.intermediate_size, H))
        self.b1 = nn.Parameter(torch.empty(d.intermediate_size))
        self.w2 = nn.Parameter(torch.empty(H, d.intermediate_size))
        self.b2 = nn.Parameter(torch.empty(H))


class _RotaryMixin:
    def _rope(self, device):
        if getattr(self, "_rope_cache", None) is None or \
                self._rope_cache[0].device != device:
            cs, sn = torch_ref.rope_tables(
                self.dims.max_position_embeddings, self.dims.head_dim,
                10000.0, device=device,
            )
            self._rope_cache = (cs, sn)
        return self._rope_cache


class Esm2Encoder(nn.Module, _RotaryMixin):
    def __init__(self, dims: ModelDims, dtype=torch.bfloat16,
                 token_dropout: bool = True):
        super().__init__()
        self.dims = dims
        self.token_dropout = token_dropout
        H = dims.hidden_size
        self.word_emb = nn.Parameter(torch.empty(dims.vocab_size, H))
        self.layers = nn.ModuleList(
            Esm2Layer(dims) for _ in range(dims.num_layers)
        )
        self.final_ln_w = nn.Parameter(torch.empty(H))
        self.final_ln_b = nn.Parameter(torch.empty(H))
        self.reset_parameters()
        self.to(dtype)
        self._rope_cache = None

    @torch.no_grad()
    def reset_parameters(self):
        for n, p in self.named_parameters():
            if p.dim() >= 2:
                nn.init.normal_(p, std=0.02)
            elif n.endswith("_w"):
                nn.init.ones_(p)
            else:
                nn.init.zeros_(p)

    @property
    def device(self):
        return self.word_emb.device

    @property
    def dtype(self):
        return self.word_emb.dtype

    @property
    def embedding_size(self) -> int:
        return self.dims.hidden_size

    @torch.no_grad()
    def forward(self, input_ids, attention_mask):
        d = self.dims
        ids = input_ids.long()
        B, S = ids.shape
        x = self.word_emb[ids]
        if (x.is_cuda and ops.HAVE_HIP and attention_mask is not None
                and not os.environ.get("DISTLLM_NO_VARLEN")
                and (os.environ.get("DISTLLM_FORCE_VARLEN")
                     or (float(attention_mask.float().mean()) < 0.7
                         and float(attention_mask.float().sum())
                         >= 256 * B))):
            # packed pays when padding is substantial AND sequences are
            # long enough to amortize the per-layer pack overheads:
            # short-buffer semantic-chunk batches (mean ~190 tokens)
            # measured 73 docs/s padded vs 46.6 packed, while the
            # length-dispersed protein batches (mean 344 of 1024)
            # measured 2.66x FASTER packed (profiles/RESULTS.md)
            return self._forward_packed(ids, attention_mask, x)
        if self.token_dropout:
            masked = ids == ESM_MASK_TOKEN
            x = x.masked_fill(masked.unsqueeze(-1), 0.0)
            mask_ratio_train = 0.15 * 0.8
            src_len = attention_mask.sum(-1, keepdim=True).float()
            ratio_obs = masked.float().sum(-1, keepdim=True) / src_len.clamp(
                min=1.0
            )
            scale = (1 - mask_ratio_train) / (1 - ratio_obs).clamp(min=1e-6)
            x = (x.float() * scale.unsqueeze(-1)).to(x.dtype)
        mask_f = torch.where(
            attention_mask.bool(),
            torch.zeros((), device=ids.device, dtype=torch.float32),
            torch.full((), float("-inf"), device=ids.device,
                       dtype=torch.float32),
        )
        pos = (
            torch.arange(S, device=ids.device, dtype=torch.int32)
            .repeat(B, 1)
            .reshape(-1)
        )
        cs, sn = self._rope(x.device)
        nh, hd = d.num_heads, d.head_dim
        H = d.hidden_size
        for layer in self.layers:
            h = ops.layer_norm(x, layer.ln1_w, layer.ln1_b,
                               eps=d.layer_norm_eps)
            qkv = ops.linear(h, layer.wqkv, layer.bqkv)
            q = qkv[..., :H].view(B, S, nh, hd)
            k = qkv[..., H : 2 * H].view(B, S, nh, hd)
            v = qkv[..., 2 * H :].view(B, S, nh, hd)
            q = ops.rope_apply(q, cs, sn, pos)
            k = ops.rope_apply(k, cs, sn, pos)
            ctx = ops.attn_fwd(q, k, v, mask=mask_f, causal=False,
                               scale=1.0 / math.sqrt(hd))
            x = x + ops.linear(ctx.reshape(B, S, H), layer.wo, layer.bo)
            h = ops.layer_norm(x, layer.ln2_w, layer.ln2_b,
                               eps=d.layer_norm_eps)
            act = ops.ACT_GELU_ERF
            ff = ops.linear(h, layer.w1, layer.b1, act=act)
            x = x + ops.linear(ff, layer.w2, layer.b2)
        return ops.layer_norm(x, self.final_ln_w, self.final_ln_b,
                              eps=d.layer_norm_eps)

    @torch.no_grad()
    def _forward_packed(self, ids, attention_mask, x):
        """Varlen path for mixed-length protein batches: pack the valid
        tokens ([T, H]), run every GEMM/norm on packed rows (zero padded
        FLOPs) and attention through the cu_seqlens flash kernel
        (reference used faesm flash-attn varlen,
        distllm/embed/encoders/esm2.py:41-55). Returns padded [B,S,H]
        with zeros at padding (poolers mask them out)."""
        d = self.dims
        B, S = ids.shape
        maskb = attention_mask.bool()
        lens = attention_mask.int().sum(1)
        cu = torch.zeros(B + 1, dtype=torch.int32, device=x.device)
        cu[1:] = lens.cumsum(0)
        pos = (attention_mask.int().cumsum(1) - 1).clamp(min=0)[maskb]
        bmap = ops.varlen_block_map(cu)
        xp = x[maskb]  # [T, H]
        if self.token_dropout:
            masked = (ids == ESM_MASK_TOKEN)[maskb]
            xp = xp.masked_fill(masked.unsqueeze(-1), 0.0)
            mask_ratio_train = 0.15 * 0.8
            # per-sequence observed-mask ratio, broadcast to tokens
            seg = torch.repeat_interleave(
                torch.arange(B, device=x.device), lens.long())
            nmask = torch.zeros(B, device=x.device).index_add_(
                0, seg, masked.float())
            ratio = nmask / lens.float().clamp(min=1.0)
            scale = (1 - mask_ratio_train) / (1 - ratio).clamp(min=1e-6)
            xp = (xp.float() * scale[seg].unsqueeze(-1)).to(xp.dtype)
        cs, sn = self._rope(x.device)
        nh, hd = d.num_heads, d.head_dim
        H = d.hidden_size
        T0 = xp.shape[0]
        Tp = (T0 + 255) // 256 * 256
        if Tp != T0:  # keep GEMMs on autotuned 256-aligned shapes
            xp = torch.nn.functional.pad(xp, (0, 0, 0, Tp - T0))
            pos = torch.nn.functional.pad(pos, (0, Tp - T0))
        T = xp.shape[0]
        for layer in self.layers:
            h = ops.layer_norm(xp, layer.ln1_w, layer.ln1_b,
                               eps=d.layer_norm_eps)
            qkv = ops.linear(h, layer.wqkv, layer.bqkv)
            q = qkv[..., :H].view(T, nh, hd)
            k = qkv[..., H: 2 * H].view(T, nh, hd)
            v = qkv[..., 2 * H:].view(T, nh, hd).contiguous()
            q = ops.rope_apply(q.contiguous(), cs, sn, pos)
            k = ops.rope_apply(k.contiguous(), cs, sn, pos)
            ctx = ops.attn_varlen(q, k, v, cu, bmap,
                                  scale=1.0 / math.sqrt(hd))
            xp = xp + ops.linear(ctx.reshape(T, H), layer.wo, layer.bo)
            h = ops.layer_norm(xp, layer.ln2_w, layer.ln2_b,
                               eps=d.layer_norm_eps)
            ff = ops.linear(h, layer.w1, layer.b1, act=ops.ACT_GELU_ERF)
            xp = xp + ops.linear(ff, layer.w2, layer.b2)
        xp = ops.layer_norm(xp, self.final_ln_w, self.final_ln_b,
                            eps=d.layer_norm_eps)
        out = torch.zeros(B, S, H, dtype=xp.dtype, device=xp.device)
        out[maskb] = xp[:T0]
        return out


class EsmCLayer(nn.Module):
    def __init__(self, d: ModelDims):
        super().__init__()
        H = d.hidden_size
        self.ln1_w = nn.Parameter(torch.empty(H))
        self.ln1_b = nn.Parameter(torch.empty(H))
        self.wqkv = nn.Parameter(torch.empty(3 * H, H))
        self.wo = nn.Parameter(torch.empty(H, H))
        self.ln2_w = nn.Parameter(torch.empty(H))
        self.ln2_b = nn.Parameter(torch.empty(H))
        self.wgate = nn.Parameter(torch.empty(d.intermediate_size, H))
        self.wup = nn.Parameter(torch.empty(d.intermediate_size, H))
        self.wdown = nn.Parameter(torch.empty(H, d.intermediate_size))


class EsmCEncoder(nn.Module, _RotaryMixin):
    """ESMC-style: pre-LN, rotary, SwiGLU FFN, residue scaling."""

    def __init__(self, dims: ModelDims, dtype=torch.bfloat16):
        super().__init__()
        self.dims = dims
        H = dims.hidden_size
        self.word_emb = nn.Parameter(torch.empty(dims.vocab_size, H))
        self.layers = nn.ModuleList(
            EsmCLayer(dims) for _ in range(dims.num_layers)
        )
        self.final_ln_w = nn.Parameter(torch.empty(H))
        self.final_ln_b = nn.Parameter(torch.empty(H))
        self.residue_scale = 1.0 / math.sqrt(dims.num_layers / 36.0)
        self.reset_parameters()
        self.to(dtype)
        self._rope_cache = None

    reset_parameters = Esm2Encoder.reset_parameters
    device = Esm2Encoder.device
    dtype = Esm2Encoder.dtype
    embedding_size = Esm2Encoder.embedding_size

    @torch.no_grad()
    def forward(self, input_ids, attention_mask):
        d = self.dims
        ids = input_ids.long()
        B, S = ids.shape
        x = self.word_emb[ids]
        if (x.is_cuda and ops.HAVE_HIP and attention_mask is not None
                and not os.environ.get("DISTLLM_NO_VARLEN")
                and (os.environ.get("DISTLLM_FORCE_VARLEN")
                     or (float(attention_mask.float().mean()) < 0.7
                         and float(attention_mask.float().sum())
                         >= 256 * B))):
            # packed pays when padding is substantial AND sequences are
            # long enough to amortize the per-layer pack overheads:
            # short-buffer semantic-chunk batches (mean ~190 tokens)
            # measured 73 docs/s padded vs 46.6 packed, while the
            # length-dispersed protein batches (mean 344 of 1024)
            # measured 2.66x FASTER packed (profiles/RESULTS.md)
            return self._forward_packed(ids, attention_mask, x)
        mask_f = torch.where(
            attention_mask.bool(),
            torch.zeros((), device=ids.device, dtype=torch.float32),
            torch.full((), float("-inf"), device=ids.device,
                       dtype=torch.float32),
        )
        pos = (
            torch.arange(S, device=ids.device, dtype=torch.int32)
            .repeat(B, 1)
            .reshape(-1)
        )
        cs, sn = self._rope(x.device)
        nh, hd = d.num_heads, d.head_dim
        H = d.hidden_size
        rs = self.residue_scale
        for layer in self.layers:
            h = ops.layer_norm(x, layer.ln1_w, layer.ln1_b,
                               eps=d.layer_norm_eps)
            qkv = ops.linear(h, layer.wqkv)
            q = qkv[..., :H].view(B, S, nh, hd)
            k = qkv[..., H : 2 * H].view(B, S, nh, hd)
            v = qkv[..., 2 * H :].view(B, S, nh, hd)
            q = ops.rope_apply(q, cs, sn, pos)
            k = ops.rope_apply(k, cs, sn, pos)
            ctx = ops.attn_fwd(q, k, v, mask=mask_f, causal=False,
                               scale=1.0 / math.sqrt(hd))
            x = x + ops.linear(ctx.reshape(B, S, H), layer.wo) * rs
            h = ops.layer_norm(x, layer.ln2_w, layer.ln2_b,
                               eps=d.layer_norm_eps)
            gate = ops.linear(h, layer.wgate)
            up = ops.linear(h, layer.wup)
            x = x + ops.linear(ops.swiglu(gate, up), layer.wdown) * rs
        return ops.layer_norm(x, self.final_ln_w, self.final_ln_b,
                              eps=d.layer_norm_eps)

    @torch.no_grad()
    def _forward_packed(self, ids, attention_mask, x):
        """Varlen path (see Esm2Encoder._forward_packed)."""
        d = self.dims
        B, S = ids.shape
        maskb = attention_mask.bool()
        lens = attention_mask.int().sum(1)
        cu = torch.zeros(B + 1, dtype=torch.int32, device=x.device)
        cu[1:] = lens.cumsum(0)
        pos = (attention_mask.int().cumsum(1) - 1).clamp(min=0)[maskb]
        bmap = ops.varlen_block_map(cu)
        xp = x[maskb]
        cs, sn = self._rope(x.device)
        nh, hd = d.num_heads, d.head_dim
        H = d.hidden_size
        T0 = xp.shape[0]
        Tp = (T0 + 255) // 256 * 256
        if Tp != T0:
            xp = torch.nn.functional.pad(xp, (0, 0, 0, Tp - T0))
            pos = torch.nn.functional.pad(pos, (0, Tp - T0))
        T = xp.shape[0]
        rs = self.residue_scale
        for layer in self.layers:
            h = ops.layer_norm(xp, layer.ln1_w, layer.ln1_b,
                               eps=d.layer_norm_eps)
            qkv = ops.linear(h, layer.wqkv)
            q = qkv[..., :H].view(T, nh, hd)
            k = qkv[..., H: 2 * H].view(T, nh, hd)
            v = qkv[..., 2 * H:].view(T, nh, hd).contiguous()
            q = ops.rope_apply(q.contiguous(), cs, sn, pos)
            k = ops.rope_apply(k.contiguous(), cs, sn, pos)
            ctx = ops.attn_varlen(q, k, v, cu, bmap,
                                  scale=1.0 / math.sqrt(hd))
            xp = xp + ops.linear(ctx.reshape(T, H), layer.wo) * rs
            h = ops.layer_norm(xp, layer.ln2_w, layer.ln2_b,
                               eps=d.layer_norm_eps)
            gate = ops.linear(h, layer.wgate)
            up = ops.linear(h, layer.wup)
            xp = xp + ops.linear(ops.swiglu(gate, up), layer.wdown) * rs
        xp = ops.layer_norm(xp, self.final_ln_w, self.final_ln_b,
                            eps=d.layer_norm_eps)
        out = torch.zeros(B, S, H, dtype=xp.dtype, device=xp.device)
        out[maskb] = xp[:T0]
        return out
