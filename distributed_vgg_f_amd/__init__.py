"""distributed_vgg_f_amd — an MI355X-native distributed CNN training framework.

A from-scratch rebuild of the capabilities of edwhere/Distributed-VGG-F
(a TensorFlow-1.x parameter-server VGG-F trainer; see SURVEY.md — the
reference mount was empty, so parity targets cite SURVEY.md/BASELINE.json
rather than reference file:line) designed MI355X-first:

- hand-written CDNA4 (gfx950) HIP kernels for every hot op: implicit-GEMM
  MFMA convolutions, MFMA FC GEMMs, LRN, maxpool, fused softmax-xent,
  fused SGD+momentum
- one process per GPU; data parallelism via bucketed RCCL all-reduce over
  xGMI, overlapped with backward (replacing the
  reference's gRPC parameter-server sync)
- NHWC activations / HWIO weights throughout (chosen for CDNA4: LRN channel
  windows and implicit-GEMM K are lane-contiguous)
- bf16 compute with fp32 master weights in contiguous arenas

CPU tensors run a plain PyTorch fp32 reference path (the numerics oracle in
tests/); CUDA tensors require the compiled HIP extension and fail loudly if
it is missing — there is no silent eager fallback on GPU.
"""

__version__ = "0.1.0"

from .config import Config  # noqa: F401
