"""Mistral/Llama-class decoder on the distllm-amd HIP ops.

One implementation serves both reference roles:
  - SFR-Embedding-Mistral-style *embedder* (last_hidden_state +
    last-token pooling; reference encoders/auto.py used with
    examples/embed/leonardo-scaling/sfr-mistral.*.yaml), and
  - Llama/Mistral *causal LM* for the generate path (reference delegated
    to vLLM, generate/generators/vllm_backend.py; here: HIP paged-KV
    decode + fused sampling).

Pre-RMSNorm, RoPE (half-split), GQA, SwiGLU; bf16 weights, fp32 accum.
"""

from __future__ import annotations

import math

import torch
from torch import nn

from .. import ops
from ..ops import torch_ref
from .config import ModelDims


class DecoderLayer(nn.Module):
    """Column-parallel fused qkv & gate|up, row-parallel wo & wdown
    (Megatron-style sharding; all-reduce after wo and wdown). The q/k/v
    and gate/up projections live in ONE parameter each so the decode
    step issues ONE GEMM for qkv and ONE for the MLP up-path
    (7 -> 4 weight streams per layer); wq/wk/... remain
    available as row-slice views for checkpoint export and tests."""

    def __init__(self, d: ModelDims, tp: int = 1):
        super().__init__()
        H = d.hidden_size
        self.d = d
        kvh = d.kv_heads // tp
        nh = d.num_heads // tp
        hd = d.head_dim
        inter = d.intermediate_size // tp
        self._nq, self._nkv = nh * hd, kvh * hd
        self._inter = inter
        self.wqkv = nn.Parameter(torch.empty((nh + 2 * kvh) * hd, H))
        # Qwen2-family: bias on the q/k/v projections only
        self.bqkv = (nn.Parameter(torch.empty((nh + 2 * kvh) * hd))
                     if d.attention_bias else None)
        self.wo = nn.Parameter(torch.empty(H, nh * hd))
        self.in_norm = nn.Parameter(torch.empty(H))
        self.post_norm = nn.Parameter(torch.empty(H))
        self.wgup = nn.Parameter(torch.empty(2 * inter, H))
        self.wdown = nn.Parameter(torch.empty(H, inter))

    @property
    def wq(self):
        return self.wqkv[: self._nq]

    @property
    def wk(self):
        return self.wqkv[self._nq: self._nq + self._nkv]

    @property
    def wv(self):
        return self.wqkv[self._nq + self._nkv:]

    @property
    def wgate(self):
        return self.wgup[: self._inter]

    @property
    def wup(self):
        return self.wgup[self._inter:]


class DecoderModel(nn.Module):
    """forward() -> last_hidden_state; generation via prefill()/decode()."""

    def __init__(self, dims: ModelDims, dtype=torch.bfloat16,
                 with_lm_head: bool | None = None,
                 tp_group=None):
        super().__init__()
        import torch.distributed as dist

        self.dims = dims
        self.tp_group = tp_group
        self.tp = (dist.get_world_size(tp_group)
                   if tp_group is not None and dist.is_initialized() else 1)
        self.tp_rank = (dist.get_rank(tp_group)
                        if tp_group is not None and dist.is_initialized()
                        else 0)
        if dims.num_heads % self.tp or dims.kv_heads % self.tp or \
                dims.intermediate_size % self.tp:
            raise ValueError(
                f"tp={self.tp} must divide heads/kv_heads/intermediate"
            )
        H = dims.hidden_size
        self.tok_emb = nn.Parameter(torch.empty(dims.vocab_size, H))
        self.layers = nn.ModuleList(
            DecoderLayer(dims, tp=self.tp) for _ in range(dims.num_layers)
        )
        self.final_norm = nn.Parameter(torch.empty(H))
        if with_lm_head is None:
            with_lm_head = dims.arch in ("llama", "qwen2")
        self.lm_head = (
            nn.Parameter(torch.empty(dims.vocab_size, H))
            if with_lm_head and not dims.tie_word_embeddings
            else None
        )
        self.with_lm_head = with_lm_head
        self.reset_parameters()
        self.to(dtype)
        self._rope_cache: tuple[torch.Tensor, torch.Tensor] | None = None
        self._fp8w: list[dict] | None = None

    @torch.no_grad()
    def enable_fp8_decode(self) -> None:
        """Opt-in W8A16 decode: quantize the projection weights to OCP
        fp8-e4m3 (per-output-channel scales) for the small-batch GEMV
        path. Prefill and B > 8 decode keep the bf16 weights — the fp8
        copies add ~0.5 byte/param. Idempotent."""
        if self._fp8w is not None:
            return
        fp8 = []
        for layer in self.layers:
            fp8.append({
                "wqkv": ops.quantize_fp8_rows(layer.wqkv),
                "wo": ops.quantize_fp8_rows(layer.wo),
                "wgup": ops.quantize_fp8_rows(layer.wgup),
                "wdown": ops.quantize_fp8_rows(layer.wdown),
            })
        self._fp8w = fp8

    @torch.no_grad()
    def reset_parameters(self):
        for n, p in self.named_parameters():
            if p.dim() >= 2:
                nn.init.normal_(p, std=0.02)
            elif n.endswith("bqkv"):
                nn.init.zeros_(p)
            else:
                nn.init.ones_(p)

    @property
    def device(self):
        return self.tok_emb.device

    @property
    def dtype(self):
        return self.tok_emb.dtype

    @property
    def embedding_size(self) -> int:
        return self.dims.hidden_size

    def _rope(self, device):
        if self._rope_cache is None or self._rope_cache[0].device != device:
            cs, sn = torch_ref.rope_tables(
                self.dims.max_position_embeddings, self.dims.head_dim,
                self.dims.rope_theta, device=device,
            )
            self._rope_cache = (cs, sn)
        return self._rope_cache

    @property
    def local_heads(self) -> int:
        return self.dims.num_heads // self.tp

    @property
    def local_kv_heads(self) -> int:
        return self.dims.kv_heads // self.tp

    def _allreduce(self, x: torch.Tensor) -> torch.Tensor:
        if self.tp > 1:
            import torch.distributed as dist

            dist.all_reduce(x, group=self.tp_group)
        return x

    def _attn_block(self, layer: DecoderLayer, h, x, mask_f, pos,
                    kv_out=None):
        """One decoder layer given h = rms_norm(x, in_norm) already
        computed (residual adds are fused into the norm kernels).
        Returns (mlp_out, x) where the caller fuses mlp_out + x into
        the next norm."""
        d = self.dims
        B, S, _ = h.shape
        nh, kvh, hd = self.local_heads, self.local_kv_heads, d.head_dim
        cs, sn = self._rope(h.device)
        # ONE fused qkv GEMM, ONE rope launch (q|k roped, v copied)
        qkv = ops.linear(h, layer.wqkv,
                         layer.bqkv).view(B, S, nh + 2 * kvh, hd)
        qkv = ops.rope_apply(qkv, cs, sn, pos, rope_heads=nh + kvh)
        q = qkv[:, :, :nh]
        k = qkv[:, :, nh: nh + kvh]
        v = qkv[:, :, nh + kvh:]
        if kv_out is not None:
            kv_out.append((k, v))
        ctx = ops.attn_fwd(q, k, v, mask=mask_f, causal=True,
                           scale=1.0 / math.sqrt(hd))
        attn = self._allreduce(
            ops.linear(ctx.reshape(B, S, nh * hd), layer.wo))
        # fused: x' = x + attn ; h2 = rms_norm(x')
        h2, x = ops.norm_fwd_sum(attn, x, layer.post_norm,
                                 eps=d.rms_norm_eps, rms=True)
        down = self._allreduce(
            ops.linear(ops.gateup_swiglu(h2, layer.wgup), layer.wdown))
        return down, x

    @torch.no_grad()
    def forward(self, input_ids: torch.Tensor, attention_mask: torch.Tensor,
                kv_out: list | None = None) -> torch.Tensor:
        ids = input_ids.long()
        B, S = ids.shape
        x = self.tok_emb[ids]
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
        d = self.dims
        h = ops.rms_norm(x, self.layers[0].in_norm, eps=d.rms_norm_eps)
        for li, layer in enumerate(self.layers):
            mlp_out, x = self._attn_block(layer, h, x, mask_f, pos, kv_out)
            next_w = (self.layers[li + 1].in_norm
                      if li + 1 < len(self.layers) else self.final_norm)
            h, x = ops.norm_fwd_sum(mlp_out, x, next_w,
                                    eps=d.rms_norm_eps, rms=True)
        return h

    @torch.no_grad()
    def logits(self, hidden: torch.Tensor) -> torch.Tensor:
        w = self.lm_head if self.lm_head is not None else self.tok_emb
        return ops.linear(hidden, w).float()

    # ---- generation-engine entry points --------------------------------
    @torch.no_grad()
    def prefill(self, input_ids, attention_mask):
        """Full prefill; returns (last_hidden [B,S,H], kv list per layer
        of (k, v) [B,S,HKV,D])."""
        kv: list = []
        hidden = self.forward(input_ids, attention_mask, kv_out=kv)
        return hidden, kv

    @torch.no_grad()
    def decode_step(self, tokens, kcaches, vcaches, block_tables, ctx_lens):
        """One decode step for B sequences.

        tokens: [B] int; caches: per-layer paged [nblk, HKV, BLK, D];
        ctx_lens: [B] int32 length INCLUDING the new token (the new
        token's kv is written at position ctx_lens-1 by this call).
        Returns logits [B, V] f32.
        """
        d = self.dims
        B = tokens.shape[0]
        nh, kvh, hd = self.local_heads, self.local_kv_heads, d.head_dim
        cs, sn = self._rope(tokens.device)
        x = self.tok_emb[tokens.long()].unsqueeze(1)  # [B,1,H]
        pos = (ctx_lens - 1).int()
        slot = pos
        # W8A16 small-batch path: fp8 weight GEMV (enable_fp8_decode)
        fp8 = (self._fp8w if self._fp8w is not None and B <= 8
               and x.is_cuda else None)
        h = ops.rms_norm(x, self.layers[0].in_norm, eps=d.rms_norm_eps)
        for li, layer in enumerate(self.layers):
            # ONE fused qkv GEMM + ONE rope (q|k roped, v copied):
            # 5 launches -> 2 on the per-token critical path
            if fp8 is not None:
                f = fp8[li]
                qkv = ops.gemv_fp8(h[:, 0], *f["wqkv"], layer.bqkv)
                qkv = qkv.view(B, 1, nh + 2 * kvh, hd)
            else:
                qkv = ops.linear(h, layer.wqkv,
                                 layer.bqkv).view(B, 1, nh + 2 * kvh, hd)
            qkv = ops.rope_apply(qkv, cs, sn, pos, rope_heads=nh + kvh)
            q = qkv[:, 0, :nh]
            k = qkv[:, 0, nh: nh + kvh].contiguous()
            v = qkv[:, 0, nh + kvh:].contiguous()
            ops.cache_put(k, v, kcaches[li], vcaches[li],
                          block_tables, slot)
            ctx = ops.paged_attn(q.contiguous(), kcaches[li],
                                 vcaches[li], block_tables, ctx_lens,
                                 scale=1.0 / math.sqrt(hd))
            if fp8 is not None:
                attn = self._allreduce(ops.gemv_fp8(
                    ctx.reshape(B, nh * hd), *f["wo"])).unsqueeze(1)
            else:
                attn = self._allreduce(
                    ops.linear(ctx.reshape(B, 1, nh * hd), layer.wo))
            h2, x = ops.norm_fwd_sum(attn, x, layer.post_norm,
                                     eps=d.rms_norm_eps, rms=True)
            if fp8 is not None:
                gu = ops.gemv_fp8(h2[:, 0], *f["wgup"])
                inter = ops._hip_ops.swiglu_packed(gu)
                down = self._allreduce(
                    ops.gemv_fp8(inter, *f["wdown"])).unsqueeze(1)
            else:
                down = self._allreduce(
                    ops.linear(ops.gateup_swiglu(h2, layer.wgup),
                               layer.wdown))
            next_w = (self.layers[li + 1].in_norm
                      if li + 1 < len(self.layers) else self.final_norm)
            h, x = ops.norm_fwd_sum(down, x, next_w,
                                    eps=d.rms_norm_eps, rms=True)
        return self.logits(h[:, 0])
