"""code2vec_amd — an MI355X-native code2vec training/inference framework.

A from-scratch reimplementation of the capabilities of sonoisa/code2vec
(reference: /root/reference) designed for AMD Instinct MI355X (gfx950):

- hand-written HIP/CDNA4 kernels for the hot path (fused embedding
  gather+concat, MFMA context-combiner GEMM with LayerNorm+tanh epilogue,
  fused single-query attention, full-vocab log-softmax+NLL, embedding
  scatter-add backward, fused multi-tensor Adam),
- data-parallel training over RCCL/xGMI (one process per GPU,
  bucketed gradient all-reduce overlapped with backward),
- a seeded, rank-aware host data pipeline replicating the reference's
  per-epoch context resampling semantics (reference
  model/dataset_builder.py:112-210),
- CLI / file-format / metric parity with the reference's main.py.
"""

__version__ = "0.1.0"
