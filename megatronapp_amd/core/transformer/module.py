"""MegatronModule base class + float16 wrapper.

Reference: megatron/core/transformer/module.py. The Float16Module keeps
bf16/fp16 parameters and casts the pipeline-boundary inputs/outputs; on
MI355X bf16 is the native training dtype (MFMA bf16 ≈ 2.5 PF dense).
"""

from __future__ import annotations

import torch
from torch import nn

from ..transformer_config import TransformerConfig


class MegatronModule(nn.Module):
    def __init__(self, config: TransformerConfig):
        super().__init__()
        self.config = config

    def state_dict_for_save_checkpoint(self, prefix="", keep_vars=False):
        return self.state_dict(prefix=prefix, keep_vars=keep_vars)

    def sharded_state_dict(self, prefix: str = "", sharded_offsets=(), metadata=None):
        from ..dist_checkpointing.mapping import module_sharded_state_dict
        return module_sharded_state_dict(self, prefix)


def conversion_helper(val, conversion):
    if isinstance(val, (tuple, list)):
        return type(val)(conversion_helper(v, conversion) for v in val)
    return conversion(val)


def fp32_to_float16(val, float16_convertor):
    def half_conversion(v):
        if isinstance(v, torch.Tensor) and v.is_floating_point() and v.dtype == torch.float32:
            return float16_convertor(v)
        return v
    return conversion_helper(val, half_conversion)


def float16_to_fp32(val):
    def float_conversion(v):
        if isinstance(v, torch.Tensor) and v.is_floating_point() and v.dtype in (
                torch.float16, torch.bfloat16):
            return v.float()
        return v
    return conversion_helper(val, float_conversion)


class Float16Module(MegatronModule):
    """Wraps a model whose parameters have been cast to fp16/bf16."""

    def __init__(self, config: TransformerConfig, module: nn.Module):
        super().__init__(config)
        self.add_module("module", module)
        if config.bf16:
            self.module = module.bfloat16()
            self.float16_convertor = lambda v: v.bfloat16()
        elif config.fp16:
            self.module = module.half()
            self.float16_convertor = lambda v: v.half()
        else:
            raise ValueError("Float16Module requires fp16 or bf16")

    def set_input_tensor(self, input_tensor):
        return self.module.set_input_tensor(input_tensor)

    def forward(self, *inputs, **kwargs):
        from .. import parallel_state
        if parallel_state.is_pipeline_first_stage():
            inputs = fp32_to_float16(inputs, self.float16_convertor)
        outputs = self.module(*inputs, **kwargs)
        if parallel_state.is_pipeline_last_stage():
            outputs = float16_to_fp32(outputs)
        return outputs

    def state_dict(self, destination=None, prefix="", keep_vars=False):
        return self.module.state_dict(destination=destination, prefix=prefix,
                                      keep_vars=keep_vars)

    def state_dict_for_save_checkpoint(self, prefix="", keep_vars=False):
        return self.module.state_dict_for_save_checkpoint(prefix, keep_vars)

    def sharded_state_dict(self, prefix: str = "", *args, **kwargs):
        return self.module.sharded_state_dict(prefix, *args, **kwargs)

    def load_state_dict(self, state_dict, strict=True):
        return self.module.load_state_dict(state_dict, strict=strict)
