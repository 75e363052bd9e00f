"""megatronapp_amd — an MI355X-native Megatron-style LLM pretraining framework.

A from-scratch reimplementation of the capabilities of OpenSQZ/MegatronApp
(Megatron-LM + MegaScan/MegaScope/MegaDPP/MegaFBD) designed MI355X-first:

* PyTorch-ROCm tensors, one process per GPU, RCCL (``torch.distributed``
  backend "nccl") over xGMI for every collective.
* Hand-written HIP/CDNA4 (gfx950) kernels for the transformer hot path
  (``megatronapp_amd.ops``): fused RMSNorm/LayerNorm, bias+SwiGLU/GeLU,
  RoPE, scaled-masked softmax, MFMA flash attention, multi-tensor Adam.
* C++ shared-memory transport for the dynamic pipeline scheduler
  (``megatronapp_amd.dpp``).

Layering mirrors the reference's concepts without copying
its code: core/parallel_state -> tensor_parallel -> transformer/models ->
pipeline_parallel + distributed + optimizer -> training.
"""

__version__ = "0.1.0"
