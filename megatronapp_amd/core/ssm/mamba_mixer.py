"""Mamba (S6) mixer (reference core/ssm/mamba_mixer.py, which wraps the
mamba-ssm CUDA/Triton package; rebuilt on the chunked torch scan + HIP
decode kernel in selective_scan.py).

Pipeline per token stream [s, b, h]:
  in_proj (column-parallel) -> x, z each [b, l, d_inner/tp]
  causal depthwise conv1d(k=4) + silu on x
  x_proj -> dt(rank), B(n), C(n);  dt_proj + softplus -> per-channel dt
  selective scan -> y;  y * silu(z);  out_proj (row-parallel) -> [s, b, h]

TP shards d_inner; B/C/dt are computed per shard from the shard's x so no
extra collectives beyond the two linear layers' own.
"""

from __future__ import annotations

import math

import torch
import torch.nn.functional as F

from .. import parallel_state
from ..tensor_parallel.layers import ColumnParallelLinear, RowParallelLinear
from ..tensor_parallel.utils import divide
from ..transformer_config import TransformerConfig
from ..transformer.module import MegatronModule
from .selective_scan import selective_scan


class MambaMixer(MegatronModule):
    def __init__(self, config: TransformerConfig, layer_number: int = 1,
                 d_state: int = 16, d_conv: int = 4, expand: int = 2,
                 dt_rank: int = None):
        super().__init__(config)
        self.layer_number = layer_number
        h = config.hidden_size
        self.d_inner = expand * h
        self.d_state = d_state
        self.d_conv = d_conv
        self.dt_rank = dt_rank or math.ceil(h / 16)
        tp = parallel_state.get_tensor_model_parallel_world_size()
        self.d_inner_local = divide(self.d_inner, tp)

        self.in_proj = ColumnParallelLinear(
            h, 2 * self.d_inner, config=config,
            init_method=config.init_method, bias=False, skip_bias_add=False,
            gather_output=False)
        self.conv1d = torch.nn.Conv1d(
            self.d_inner_local, self.d_inner_local, d_conv,
            groups=self.d_inner_local, padding=d_conv - 1, bias=True,
            dtype=config.params_dtype)
        self.x_proj = torch.nn.Linear(
            self.d_inner_local, self.dt_rank + 2 * d_state, bias=False,
            dtype=config.params_dtype)
        self.dt_proj = torch.nn.Linear(self.dt_rank, self.d_inner_local,
                                       bias=True, dtype=config.params_dtype)
        # dt bias init so softplus(dt) starts in [1e-3, 0.1] (S6 init)
        with torch.no_grad():
            dt_init = torch.exp(
                torch.rand(self.d_inner_local) *
                (math.log(0.1) - math.log(1e-3)) + math.log(1e-3))
            self.dt_proj.bias.copy_(
                (dt_init + torch.log(-torch.expm1(-dt_init))).to(
                    config.params_dtype))

        A = torch.arange(1, d_state + 1, dtype=torch.float32).repeat(
            self.d_inner_local, 1)
        self.A_log = torch.nn.Parameter(torch.log(A))
        self.D = torch.nn.Parameter(torch.ones(self.d_inner_local))
        self.out_proj = RowParallelLinear(
            self.d_inner, h, config=config,
            init_method=config.output_layer_init_method, bias=False,
            input_is_parallel=True, skip_bias_add=True)

    def forward(self, hidden_states, inference_context=None, **kwargs):
        # [s, b, h] -> [b, l, h]
        s, b, _ = hidden_states.shape
        xz, _ = self.in_proj(hidden_states)          # [s, b, 2*din/tp]
        xz = xz.transpose(0, 1)                      # [b, l, 2*din]
        x, z = xz.chunk(2, dim=-1)

        conv_state = None
        if inference_context is not None:
            conv_state, ssm_state = self._get_states(inference_context, b, x)

        # causal depthwise conv over l
        xt = x.transpose(1, 2)                       # [b, din, l]
        if conv_state is not None and s == 1:
            # decode: roll the conv window
            conv_state.copy_(torch.cat([conv_state[..., 1:], xt], dim=-1))
            xc = (conv_state * self.conv1d.weight.squeeze(1)).sum(-1)
            xc = (xc + self.conv1d.bias).unsqueeze(-1)
        else:
            xc = self.conv1d(xt)[..., :s]
            if conv_state is not None:
                pad = self.d_conv - min(self.d_conv, s)
                tail = xt[..., -self.d_conv:]
                if pad:
                    tail = F.pad(tail, (pad, 0))
                conv_state.copy_(tail)
        x = F.silu(xc.transpose(1, 2))               # [b, l, din]

        prm = self.x_proj(x)                         # [b, l, rank+2n]
        dt, B, C = torch.split(
            prm, [self.dt_rank, self.d_state, self.d_state], dim=-1)
        dt = F.softplus(self.dt_proj(dt))            # [b, l, din]
        A = -torch.exp(self.A_log.float())

        if inference_context is not None:
            y, new_state = selective_scan(
                x.contiguous(), dt.contiguous(), A, B.contiguous(),
                C.contiguous(), self.D.float(), h0=ssm_state,
                return_state=True)
            ssm_state.copy_(new_state)
        else:
            y = selective_scan(x.contiguous(), dt.contiguous(), A,
                               B.contiguous(), C.contiguous(), self.D.float())

        y = y * F.silu(z)
        y = y.transpose(0, 1)                        # [s, b, din]
        out, bias = self.out_proj(y)
        return out, bias

    def _get_states(self, ctx, b, x):
        key = ("mamba", self.layer_number)
        store = ctx.key_value_memory_dict
        if key not in store:
            conv = torch.zeros(b, self.d_inner_local, self.d_conv,
                               dtype=x.dtype, device=x.device)
            ssm = torch.zeros(b, self.d_inner_local, self.d_state,
                              dtype=torch.float32, device=x.device)
            store[key] = (conv, ssm)
        return store[key]
