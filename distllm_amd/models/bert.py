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
