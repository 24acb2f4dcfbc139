"""Beam-search decoding over the native decoder models.

Reference parity: the reference's huggingface backend passes
num_beams (default 10) to transformers model.generate
(distllm/generate/generators/huggingface_backend.py:41-44). The native
engine's hot path is sampling (the vLLM-parity path); beam search is
provided for the `huggingface`-alias semantics via full-forward steps
— all beams batched into ONE forward per step, so on GPU it runs the
same HIP kernel path as any batch-B forward.
"""

from __future__ import annotations

import torch


@torch.no_grad()
def beam_generate(model, prompt_ids: list[int], num_beams: int,
                  max_tokens: int, length_penalty: float = 1.0,
                  eos_token_id: int | None = None) -> list[int]:
    """Return the generated ids (prompt excluded) of the best beam by
    length-normalized cumulative log-probability."""
    dev = model.device
    # Clamp to the model's positional range (the sampling engine path
    # truncates prompts too): long prompts must not index RoPE tables
    # out of range mid-search.
    max_pos = int(model.dims.max_position_embeddings)
    prompt_ids = list(prompt_ids)[-(max_pos - 1):]
    max_tokens = min(max_tokens, max_pos - len(prompt_ids))
    beams: list[tuple[list[int], float]] = [(list(prompt_ids), 0.0)]
    done: list[tuple[list[int], float]] = []

    def score(ids, lp):
        n = max(1, len(ids) - len(prompt_ids))
        return lp / (n ** length_penalty)

    for _ in range(max_tokens):
        ids = torch.tensor([b[0] for b in beams], device=dev)
        mask = torch.ones_like(ids, dtype=torch.int32)
        hidden = model(ids, mask)
        logits = model.logits(hidden[:, -1]).float()
        logprobs = torch.log_softmax(logits, dim=-1)          # [B, V]
        k = min(num_beams, logprobs.shape[-1])
        top_lp, top_ix = torch.topk(logprobs, k, dim=-1)      # [B, k]
        cand: list[tuple[list[int], float]] = []
        for bi, (bids, blp) in enumerate(beams):
            for j in range(k):
                cand.append((bids + [int(top_ix[bi, j])],
                             blp + float(top_lp[bi, j])))
        cand.sort(key=lambda c: c[1], reverse=True)
        beams = []
        for bids, blp in cand:
            if eos_token_id is not None and bids[-1] == eos_token_id:
                done.append((bids[:-1], blp))
            else:
                beams.append((bids, blp))
            if len(beams) >= num_beams:
                break
        if not beams:
            break
        # early exit: the best live beam cannot beat the best finished one
        if done and score(*max(done, key=lambda d: score(*d))) >= \
                score(*max(beams, key=lambda b: score(*b))):
            break
    done.extend(beams)
    best = max(done, key=lambda d: score(*d))
    return best[0][len(prompt_ids):]
