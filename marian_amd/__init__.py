"""marian_amd — MI355X-native NMT training + decoding framework.

A from-scratch rebuild of the capabilities of tneck/marian-nmt-distributed
(a Marian NMT distribution) designed MI355X-first:

- expression-graph / auto-diff layer  -> PyTorch-ROCm autograd
  (ref: src/graph/expression_graph.{h,cpp})
- hot-path kernels (attention, LayerNorm, softmax, fused CE, top-k, Adam)
  -> hand-written HIP/CDNA4 (gfx950 MFMA + LDS) in marian_amd/ops/hip/
  (ref: src/tensors/gpu/*.cu)
- gradient sync -> RCCL over xGMI via torch.distributed + bucketed
  reduce-scatter / all-gather with ZeRO-1 optimizer sharding
  (ref: src/training/communicator_nccl.h, graph_group_sync.cpp)
- CLI surface -> marian-train / marian-decoder / marian-scorer /
  marian-server / marian-vocab / marian-conv
  (ref: src/command/*.cpp)
"""

__version__ = "0.1.0"
