"""distributedllm_amd — an MI355X-native layer-sliced LLM inference framework.

A from-scratch re-design of the capabilities of X-rayLaser/DistributedLLM
(layer-sliced pipeline inference of GGML LLaMA-family checkpoints across
compute nodes) built MI355X-first:

* transformer slice forward = hand-written CDNA4 HIP kernels (MFMA, LDS)
  with weights resident in 288 GB HBM3E per GPU,
* inter-slice activation hops = RCCL point-to-point over xGMI
  (``torch.distributed`` backend "nccl", which is RCCL on ROCm),
* host-side control plane (provisioning, uploads, status) = a TCP protocol
  of the same shape as the reference's (distllm/protocol.py), re-designed.

Reference capability map: see SURVEY.md at the repo root.
"""

__version__ = "0.1.0"
