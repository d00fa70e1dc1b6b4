"""nanorlhf_amd — a brand-new MI355X-native single-node RLHF framework.

Capabilities mirror jackfsuia/nanoRLHF (GRPO / PPO / RLOO / ReMax / RAFT /
REINFORCE + sparse-GRPO "r1" mode), re-designed MI355X-first:

  * rollout generation is an in-process paged-KV sampler (hand-written CDNA4
    HIP kernels: MFMA prefill attention, LDS-staged paged decode attention,
    fused RMSNorm/RoPE/SwiGLU, temperature/top-p sampling) instead of the
    reference's boot-vLLM-from-a-merged-checkpoint-on-disk round trip
    (reference: GRPO/grpo_trainer.py:122-166),
  * the policy-update hot path (fused token-logprob/entropy over the 151k
    vocab, masked whitening, fused AdamW) is hand-written HIP for gfx950,
  * reference/reward/value models offload to pinned host memory on a side
    stream, sized against 288 GB HBM3E per GPU (most stay resident),
  * DP scaling uses RCCL over xGMI, one process per GPU, gradient all-reduce
    overlapped with backward.

No CUDA shims, no Triton, no hipify: the compute path is PyTorch-ROCm +
hand-written HIP/CDNA4 kernels + RCCL.
"""

__version__ = "0.1.0"
