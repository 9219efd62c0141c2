"""pytorch_distributed_train_amd — MI355X-native distributed data-parallel training.

A from-scratch rebuild of the capabilities of zoezhu/pytorch_distributed_train
(the reference mount was empty, so the capability contract is
BASELINE.json) designed MI355X-first:

- one process per GPU, RCCL collectives over xGMI (our own C++ extension owns
  the ncclComm_t; gloo via torch.distributed only for CPU plumbing tests),
- hand-written CDNA4 HIP kernels (gfx950, MFMA/LDS-tiled) for the hot path:
  implicit-GEMM conv, BatchNorm/SyncBatchNorm, fused softmax-cross-entropy,
  fused SGD+momentum, MFMA GEMM for linear/attention,
- NHWC activation layout in bf16/fp16, fp32 master weights stored flat so the
  optimizer is one kernel over one buffer,
- a DDP gradient-bucket reducer whose buckets are contiguous slices of a flat
  gradient buffer, all-reduced on a side HIP stream overlapped with backward.

Layer map mirrors SURVEY.md §1: launcher -> rendezvous -> communicator ->
DDP reducer -> models/ops -> kernels.
"""

__version__ = "0.1.0"
