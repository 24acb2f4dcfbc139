"""BERT-class bidirectional encoder, built on the distllm-amd HIP ops.

Replaces the reference's HF AutoModel path for S-PubMedBERT-class
encoders (reference: distllm/embed/encoders/auto.py:34-138). Forward
semantics match HF BertModel (post-LayerNorm, absolute position
embeddings, erf-GELU) so pretrained weights can be mapped in; compute
runs through the hand-written CDNA4 kernels (attention, fused
residual-LayerNorm, fused GELU) on GPU.

Inference-only, bf16 weights, fp32 accumulation inside kernels.
"""

from __future__ import annotations

import math
import os

import torch
from torch import nn

from .. import ops
from .config import ModelDims


class BertLayer(nn.Module):
    def __init__(self, d: ModelDims):
        super().__init__()
        H = d.hidden_size
        self.d = d
        self.wqkv = nn.Parameter(torch.empty(3 * H, H))
        self.bqkv = nn.Parameter(torch.empty(3 * H))
        self.wo = nn.Parameter(torch.empty(H, H))
        self.bo = nn.Parameter(torch.empty(H))
        self.ln1_w = nn.Parameter(torch.empty(H))
        self.ln1_b = nn.Parameter(torch.empty(H))
        self.w1 = nn.Parameter(torch.empty(d.intermediate_size, H))
        self.b1 = nn.Parameter(torch.empty(d.intermediate_size))
        self.w2 = nn.Parameter(torch.empty(H, d.intermediate_size))
        self.b2 = nn.Parameter(torch.empty(H))
        self.ln2_w = nn.Parameter(torch.empty(H))
        self.ln2_b = nn.Parameter(torch.empty(H))

    def forward(self, x: torch.Tensor, mask_f: torch.Tensor | None):
        d = self.d
        B, S, H = x.shape
        nh, hd = d.num_heads, d.head_dim
        qkv = ops.linear(x, self.wqkv, self.bqkv)  # [B,S,3H]
        q = qkv[..., :H].view(B, S, nh, hd)
        k = qkv[..., H : 2 * H].view(B, S, nh, hd)
        v = qkv[..., 2 * H :].view(B, S, nh, hd)
        ctx = ops.attn_fwd(q, k, v, mask=mask_f, causal=False,
                           scale=1.0 / math.sqrt(hd))
        attn_out = ops.linear(ctx.reshape(B, S, H), self.wo, self.bo)
        x = ops.layer_norm(attn_out, self.ln1_w, self.ln1_b,
                           eps=d.layer_norm_eps, res=x)
        act = ops.ACT_GELU_ERF if d.gelu == "erf" else ops.ACT_GELU_TANH
        h = ops.linear(x, self.w1, self.b1, act=act)
        out = ops.linear(h, self.w2, self.b2)
        x = ops.layer_norm(out, self.ln2_w, self.ln2_b,
                           eps=d.layer_norm_eps, res=x)
        return x


class BertEncoder(nn.Module):
    """HF-BertModel-compatible forward: (input_ids, attention_mask,
    token_type_ids?) -> last_hidden_state [B, S, H]."""

    def __init__(self, dims: ModelDims, dtype=torch.bfloat16):
        super().__init__()
        self.dims = dims
        H = dims.hidden_size
        self.word_emb = nn.Parameter(torch.empty(dims.vocab_size, H))
        self.pos_emb = nn.Parameter(
            torch.empty(dims.max_position_embeddings, H)
        )
        self.type_emb = nn.Parameter(torch.empty(dims.type_vocab_size, H))
        self.emb_ln_w = nn.Parameter(torch.empty(H))
        self.emb_ln_b = nn.Parameter(torch.empty(H))
        self.layers = nn.ModuleList(
            BertLayer(dims) for _ in range(dims.num_layers)
        )
        self.reset_parameters()
        self.to(dtype)

    @torch.no_grad()
    def reset_parameters(self):
        for n, p in self.named_parameters():
            if p.dim() >= 2:
                nn.init.normal_(p, std=0.02)
            elif n.endswith(("ln_w", "ln1_w", "ln2_w")):
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
    def forward(
        self,
        input_ids: torch.Tensor,
        attention_mask: torch.Tensor,
        token_type_ids: torch.Tensor | None = None,
    ) -> torch.Tensor:

        ids = input_ids.long()
        B, S = ids.shape
        # BERT packed/varlen is opt-in only: on its realistic padded
        # workloads (semantic-chunk buffers AND chunk re-embeds, 12
        # light layers, ~200-400-token rows) the padded path measured
        # FASTER (73 vs 47-50 docs/s on the pipeline bench) — the
        # per-layer pack overheads beat the skipped pad FLOPs at this
        # depth/width. The deep/wide ESM encoders auto-pack instead
        # (x3.0 measured, profiles/RESULTS.md).
        if (ids.is_cuda and ops.HAVE_HIP and attention_mask is not None
                and token_type_ids is None
                and os.environ.get("DISTLLM_FORCE_VARLEN")):
            return self._forward_packed(ids, attention_mask)
        if self.dims.position_style == "roberta":
            pad = self.dims.pad_token_id
            nz = (ids != pad).int()
            pos2d = (nz.cumsum(1) * nz + pad).long()
            x = self.word_emb[ids] + self.pos_emb[pos2d]
        else:
            pos = torch.arange(S, device=ids.device)
            x = self.word_emb[ids] + self.pos_emb[pos][None, :, :]
        if token_type_ids is not None:
            x = x + self.type_emb[token_type_ids.long()]
        else:
            x = x + self.type_emb[0]
        x = ops.layer_norm(x, self.emb_ln_w, self.emb_ln_b,
                           eps=self.dims.layer_norm_eps)
        mask_f = torch.where(
            attention_mask.bool(),
            torch.zeros((), device=ids.device, dtype=torch.float32),
            torch.full((), float("-inf"), device=ids.device,
                       dtype=torch.float32),
        )
        for layer in self.layers:
            x = layer(x, mask_f)
        return x

    @torch.no_grad()
    def _forward_packed(self, ids, attention_mask):
        """Varlen path for padded text batches (jsonl_chunk buffers are
        length-dispersed): valid tokens packed to [T, H], GEMMs/norms on
        packed rows, attention through the cu_seqlens flash kernel
        (same design as the ESM packed path). Returns
        padded [B, S, H] with zeros at padding."""
        d = self.dims
        B, S = ids.shape
        maskb = attention_mask.bool()
        lens = attention_mask.int().sum(1)
        cu = torch.zeros(B + 1, dtype=torch.int32, device=ids.device)
        cu[1:] = lens.cumsum(0)
        if d.position_style == "roberta":
            nz = (ids != d.pad_token_id).int()
            pos = (nz.cumsum(1) * nz + d.pad_token_id)[maskb]
        else:
            pos = (attention_mask.int().cumsum(1) - 1).clamp(min=0)[maskb]
        bmap = ops.varlen_block_map(cu)
        x = (self.word_emb[ids[maskb]] + self.pos_emb[pos.long()]
             + self.type_emb[0])
        # pad the packed rows to a multiple of 256 so the projection
        # GEMMs keep their autotuned fused-epilogue shapes; pad rows
        # are inert (not in the block map; attn_varlen zeroes them)
        T0 = x.shape[0]
        Tp = (T0 + 255) // 256 * 256
        if Tp != T0:
            x = torch.nn.functional.pad(x, (0, 0, 0, Tp - T0))
        x = ops.layer_norm(x, self.emb_ln_w, self.emb_ln_b,
                           eps=d.layer_norm_eps)
        nh, hd = d.num_heads, d.head_dim
        H = d.hidden_size
        T = x.shape[0]
        scale = 1.0 / math.sqrt(hd)
        for layer in self.layers:
            qkv = ops.linear(x, layer.wqkv, layer.bqkv)
            q = qkv[..., :H].view(T, nh, hd).contiguous()
            k = qkv[..., H: 2 * H].view(T, nh, hd).contiguous()
            v = qkv[..., 2 * H:].view(T, nh, hd).contiguous()
            ctx = ops.attn_varlen(q, k, v, cu, bmap, scale=scale)
            attn_out = ops.linear(ctx.reshape(T, H), layer.wo, layer.bo)
            x = ops.layer_norm(attn_out, layer.ln1_w, layer.ln1_b,
                               eps=d.layer_norm_eps, res=x)
            act = ops.ACT_GELU_ERF if d.gelu == "erf" else ops.ACT_GELU_TANH
            hmid = ops.linear(x, layer.w1, layer.b1, act=act)
            out = ops.linear(hmid, layer.w2, layer.b2)
            x = ops.layer_norm(out, layer.ln2_w, layer.ln2_b,
                               eps=d.layer_norm_eps, res=x)
        padded = torch.zeros(B, S, H, dtype=x.dtype, device=x.device)
        padded[maskb] = x[:T0]
        return padded
