"""Model/parallel configuration dataclasses.

Compact equivalents of the reference's ``ModelParallelConfig``
(model_parallel_config.py) and ``TransformerConfig``
(transformer_config.py:1-1125) — only knobs the MI355X framework
implements are kept; every field name matches the reference so configs
translate 1:1.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Callable, Optional

import torch


@dataclass
class ModelParallelConfig:
    # ---- parallelism sizes ----
    tensor_model_parallel_size: int = 1
    pipeline_model_parallel_size: int = 1
    virtual_pipeline_model_parallel_size: Optional[int] = None
    context_parallel_size: int = 1
    expert_model_parallel_size: int = 1
    sequence_parallel: bool = False
    # context-parallel attention communication: "p2p" (ring attention over
    # xGMI neighbours) or "a2a" (Ulysses head scatter); SURVEY.md §5.7.
    cp_comm_type: str = "p2p"

    # ---- dtypes ----
    fp16: bool = False
    bf16: bool = False
    params_dtype: torch.dtype = torch.float32

    # ---- perf knobs ----
    gradient_accumulation_fusion: bool = True
    async_tensor_model_parallel_allreduce: bool = True
    overlap_p2p_comm: bool = True
    batch_p2p_comm: bool = True
    deallocate_pipeline_outputs: bool = True
    # depth-vs-breadth-first interleaved schedule knob (MegaDPP lever;
    # reference model_parallel_config.py:304)
    microbatch_group_size_per_vp_stage: Optional[int] = None

    # ---- runtime ----
    pipeline_dtype: Optional[torch.dtype] = None
    variable_seq_lengths: bool = False
    timers: Optional[object] = None
    finalize_model_grads_func: Optional[Callable] = None
    grad_scale_func: Optional[Callable] = None
    no_sync_func: Optional[Callable] = None
    param_sync_func: Optional[Callable] = None

    def __post_init__(self):
        if self.sequence_parallel and self.tensor_model_parallel_size == 1:
            self.sequence_parallel = False
        if self.bf16:
            self.params_dtype = torch.bfloat16
            self.pipeline_dtype = self.pipeline_dtype or torch.bfloat16
        elif self.fp16:
            self.params_dtype = torch.float16
            self.pipeline_dtype = self.pipeline_dtype or torch.float16


def _rsqrt_hidden(config: "TransformerConfig") -> float:
    return config.hidden_size ** -0.5


@dataclass
class TransformerConfig(ModelParallelConfig):
    # ---- architecture ----
    num_layers: int = 0
    hidden_size: int = 0
    num_attention_heads: int = 0
    num_query_groups: Optional[int] = None      # GQA; None -> MHA
    kv_channels: Optional[int] = None
    ffn_hidden_size: Optional[int] = None
    hidden_dropout: float = 0.1
    attention_dropout: float = 0.1
    layernorm_epsilon: float = 1e-5
    layernorm_zero_centered_gamma: bool = False
    normalization: str = "LayerNorm"             # "LayerNorm" | "RMSNorm"
    activation_func: str = "gelu"                # "gelu" | "swiglu" | "squared_relu"
    gated_linear_unit: bool = False
    add_bias_linear: bool = True
    add_qkv_bias: bool = False
    qk_layernorm: bool = False
    rotary_percent: float = 1.0
    rotary_base: int = 10000
    position_embedding_type: str = "learned_absolute"  # | "rope" | "none"
    untie_embeddings_and_output_weights: bool = False
    apply_residual_connection_post_layernorm: bool = False
    attention_softmax_in_fp32: bool = True
    apply_query_key_layer_scaling: bool = False
    masked_softmax_fusion: bool = True
    bias_activation_fusion: bool = True
    bias_dropout_fusion: bool = True
    persist_layer_norm: bool = True
    init_method_std: float = 0.02
    init_method: Optional[Callable] = None
    output_layer_init_method: Optional[Callable] = None
    apply_rope_fusion: bool = True
    window_size: Optional[tuple] = None

    # ---- MoE ----
    # multi-token prediction (DeepSeek-style; reference
    # transformer_config.py mtp section)
    mtp_num_layers: Optional[int] = None
    mtp_loss_scaling_factor: float = 0.1

    num_moe_experts: Optional[int] = None
    moe_router_topk: int = 2
    moe_router_load_balancing_type: str = "aux_loss"  # | "sinkhorn" | "none"
    # renormalize the selected top-k probs to sum to 1 (Mixtral-style
    # routing; reference moe_router_topk_scaling / norm_topk_prob)
    moe_router_renormalize: bool = False
    moe_aux_loss_coeff: float = 0.0
    moe_z_loss_coeff: Optional[float] = None
    moe_token_dispatcher_type: str = "alltoall"   # | "allgather"
    moe_grouped_gemm: bool = True
    moe_shared_expert_intermediate_size: Optional[int] = None
    moe_ffn_hidden_size: Optional[int] = None
    moe_layer_freq: int = 1

    # ---- activation recompute ----
    recompute_granularity: Optional[str] = None   # "selective" | "full"
    recompute_method: Optional[str] = None        # "uniform" | "block"
    recompute_num_layers: Optional[int] = None
    distribute_saved_activations: bool = False

    # ---- attention backend ----
    # "flash"  -> hand-written MFMA flash-attention HIP kernel (default on GPU)
    # "fused"  -> baddbmm + fused-softmax HIP kernel + bmm (reference local path)
    # "unfused"-> plain torch ops (CPU / debugging)
    attention_backend: str = "auto"
    flash_decode: bool = False

    # ---- misc ----
    fp32_residual_connection: bool = False
    clone_scatter_output_in_embedding: bool = True
    tp_comm_overlap: bool = False
    enable_cuda_graph: bool = False
    external_cuda_graph: bool = False
    calculate_per_token_loss: bool = False

    def __post_init__(self):
        super().__post_init__()
        if self.kv_channels is None and self.num_attention_heads:
            self.kv_channels = self.hidden_size // self.num_attention_heads
        if self.num_query_groups is None:
            self.num_query_groups = self.num_attention_heads
        if self.ffn_hidden_size is None:
            self.ffn_hidden_size = 4 * self.hidden_size
        if self.activation_func == "swiglu":
            self.gated_linear_unit = True
        if self.num_moe_experts is not None and self.moe_ffn_hidden_size is None:
            self.moe_ffn_hidden_size = self.ffn_hidden_size
        if self.init_method is None:
            self.init_method = _init_normal(self.init_method_std)
        if self.output_layer_init_method is None:
            self.output_layer_init_method = _init_normal(
                self.init_method_std / (2.0 * max(self.num_layers, 1)) ** 0.5)
        if self.num_attention_heads:
            assert self.num_attention_heads % self.num_query_groups == 0
        if self.recompute_granularity == "full" and self.recompute_num_layers is None:
            self.recompute_num_layers = 1


def _init_normal(std: float) -> Callable:
    def init_(tensor: torch.Tensor) -> torch.Tensor:
        return torch.nn.init.normal_(tensor, mean=0.0, std=std)
    return init_


@dataclass
class MLATransformerConfig(TransformerConfig):
    """Multi-latent-attention extension (DeepSeek-style), reference
    transformer_config.py MLA section."""
    q_lora_rank: Optional[int] = None
    kv_lora_rank: int = 512
    qk_head_dim: int = 128
    qk_pos_emb_head_dim: int = 64
    v_head_dim: int = 128
    rotary_scaling_factor: float = 1.0
    max_position_embeddings: int = 4096
