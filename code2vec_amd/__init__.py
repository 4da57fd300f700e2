"""code2vec_amd — MI355X-native path-attention code-embedding framework.

A from-scratch reimplementation of the capabilities of tech-srl/code2vec
(reference layout documented in SURVEY.md), built MI355X-first:

- PyTorch-ROCm framework layer, one model implementation (no TF/Keras dual split).
- Hot ops are hand-written CDNA4 (gfx950) HIP kernels: fused embedding
  gather+concat, MFMA-tiled transform GEMM with tanh + attention-score epilogue,
  fused masked-softmax attention reduce, fused large-vocab cross-entropy,
  dense + scatter (sparse-row) Adam.
- Data-parallel training over RCCL/xGMI (torch.distributed backend "nccl").
- Reference-compatible artifacts: `.c2v` data format, `dictionaries.bin`
  vocabulary pickles, CLI flags and checkpoint naming.
"""

__version__ = "0.1.0"
