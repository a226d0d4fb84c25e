"""MI355X-native CenterNet-style helmet detector framework.

A from-scratch rebuild of the capabilities of tyui592/Real_Time_Helmet_Detection
(reference layout surveyed in SURVEY.md) designed MI355X-first:

- PyTorch-ROCm as the tensor substrate, one process per GPU.
- Hand-written HIP/CDNA4 (gfx950) kernels for the hot ops (fused conv+BN+act,
  pooling, upsample, focal/L1 loss, peak decode + top-k, NMS) under ``ops/``.
- RCCL collectives over xGMI for data parallelism (``parallel/``), with
  gradient buckets sized for the 7-link point-to-point topology and overlapped
  with backward on a dedicated HIP stream.
- bf16 MFMA compute for AMP (``amp.py``) instead of fp16 loss-scaling.

CLI / checkpoint / export contracts follow the reference
(/root/reference/main.py, config.py, train.py; see docstrings per module).
"""

__version__ = "0.1.0"
