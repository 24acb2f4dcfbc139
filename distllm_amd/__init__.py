"""distllm-amd: MI355X-native distributed embedding, generation and RAG framework.

A from-scratch re-design of the capabilities of ramanathanlab/distllm
(reference layout: /root/reference/distllm) for AMD Instinct MI355X
(gfx950, CDNA4): PyTorch-ROCm orchestration, hand-written HIP/MFMA
kernels for the hot compute (attention, norms, activations, pooling,
kNN search, paged decode, sampling), and RCCL over xGMI for the
single-node 8-GPU data-parallel shard that replaces the reference's
Parsl task farm (reference: distllm/parsl.py).
"""

__version__ = "0.1.0"
