"""Buffer-aligned mixed-precision Adam, optionally ZeRO-1 sharded.

MI355X-first redesign of the reference's two optimizer classes
(Float16OptimizerWithFloat16Params optimizer.py:504 and
DistributedOptimizer distrib_optimizer.py:80): instead of per-param fp32
master copies and apex multi_tensor_apply over thousands of tensors, the
optimizer state lives in FLAT fp32 buffers exactly aligned with the DDP
grad/param buffers, so one fused HIP Adam kernel call per buffer updates
everything (ops/csrc/adam.hip), and the ZeRO-1 shard is just a slice of
the same flat space:

  grad buffer (fp32)  --reduce-scatter-->  local shard grads
  master (fp32 flat)  --adam kernel   -->  master shard updated
  param buffer (bf16) <--cast shard   ---  then all-gather params

With dp==1 the "shard" is the whole buffer and the collectives are no-ops,
giving one unified code path.
"""

from __future__ import annotations

import math
import warnings
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from .. import parallel_state
from ..tensor_parallel.layers import param_is_not_tensor_parallel_duplicate
from ..trace_hooks import trace_scope
from ... import ops as _ops
from .clip_grads import clip_grad_by_total_norm_fp32, count_zeros_fp32, get_grad_norm_fp32
from .optimizer_config import OptimizerConfig


def _adam_step_flat(master: torch.Tensor, grad: torch.Tensor, m: torch.Tensor,
                    v: torch.Tensor, lr: float, beta1: float, beta2: float,
                    eps: float, weight_decay: float, step: int,
                    p16: torch.Tensor = None) -> bool:
    """One Adam(W) step over a flat fp32 master/grad span; HIP kernel on
    GPU.  States may be bf16 (precision-aware optimizer) — math is fp32
    either way, bf16 states round on store.  p16 (optional): the bf16
    model-param shard, written by the kernel in the same pass (replaces
    the separate master->param cast copy).  Returns True iff the fused
    kernel ran (and consumed p16)."""
    if master.is_cuda and _ops.have_ops() and hasattr(_ops.get_ops(), "adamw_flat"):
        _ops.get_ops().adamw_flat(master, grad, m, v, lr, beta1, beta2, eps,
                                  weight_decay, step, p16)
        return True
    bias_correction1 = 1 - beta1 ** step
    bias_correction2 = 1 - beta2 ** step
    if weight_decay != 0:
        master.mul_(1 - lr * weight_decay)
    mf = m.float() if m.dtype != torch.float32 else m
    vf = v.float() if v.dtype != torch.float32 else v
    mf.mul_(beta1).add_(grad, alpha=1 - beta1)
    vf.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
    denom = (vf / bias_correction2).sqrt_().add_(eps)
    master.addcdiv_(mf, denom, value=-lr / bias_correction1)
    if mf is not m:
        m.copy_(mf)
    if vf is not v:
        v.copy_(vf)
    return False


class ParamGroup:
    """A (wd_mult, lr_mult) bucket of shard ranges within one buffer."""

    def __init__(self, wd_mult: float, lr_mult: float):
        self.wd_mult = wd_mult
        self.lr_mult = lr_mult
        self.ranges: List[tuple] = []  # (buffer_idx, start, end) in shard coords


class DistributedOptimizer:
    """Mixed-precision Adam over DDP buffers; ZeRO-1 when the DDP config
    enabled use_distributed_optimizer, full-replica otherwise."""

    def __init__(self, config: OptimizerConfig, model_chunks: List):
        self.config = config
        self.grad_scaler = None
        if getattr(config, "fp16", False):
            from .grad_scaler import ConstantGradScaler, DynamicGradScaler
            if config.loss_scale is not None:
                self.grad_scaler = ConstantGradScaler(config.loss_scale)
            else:
                self.grad_scaler = DynamicGradScaler(
                    initial_scale=config.initial_loss_scale,
                    min_scale=config.min_loss_scale,
                    growth_interval=config.loss_scale_window
                    if hasattr(config, "loss_scale_window") else 1000,
                    hysteresis=config.hysteresis
                    if hasattr(config, "hysteresis") else 2)
        self.model_chunks = model_chunks
        self.buffers = []
        for chunk in model_chunks:
            self.buffers.extend(chunk.buffers)
        self.sharded = any(b.ddp_config.use_distributed_optimizer
                           for b in self.buffers)
        self.step_count = 0
        self.lr = config.lr or 0.0
        self.weight_decay = config.weight_decay

        # per-buffer shard state
        self.shard_master: List[torch.Tensor] = []
        self.shard_m: List[torch.Tensor] = []
        self.shard_v: List[torch.Tensor] = []
        self.shard_bounds: List[tuple] = []
        # per-(buffer,param) metadata for grad-norm / wd grouping
        self.no_wd_ranges: List[List[tuple]] = []  # per buffer: shard-coord ranges w/o wd
        self.norm_ranges: List[List[tuple]] = []   # per buffer: ranges counted in grad norm

        for buf in self.buffers:
            if buf.ddp_config.use_distributed_optimizer:
                lo, hi = buf.local_shard_bounds()
            else:
                lo, hi = 0, buf.numel
            self.shard_bounds.append((lo, hi))
            if buf.param_data is not None:
                master = buf.param_data[lo:hi].float().clone()
            else:
                master = torch.empty(hi - lo, dtype=torch.float32,
                                     device=buf.grad_data.device)
                # gather initial values from the individual param tensors
                for p, (s, e) in buf.param_index_map.items():
                    os_, oe = max(s, lo), min(e, hi)
                    if os_ < oe:
                        flat = p.data.reshape(-1)
                        master[os_ - lo:oe - lo].copy_(
                            flat[os_ - s:oe - s].float())
            self.shard_master.append(master)
            st_dt = torch.float32
            if getattr(config, "use_precision_aware_optimizer", False):
                a = getattr(config, "exp_avg_dtype", "fp32")
                b = getattr(config, "exp_avg_sq_dtype", "fp32")
                if a == "bf16" and b == "bf16":
                    st_dt = torch.bfloat16
                elif "bf16" in (a, b):
                    # the fused flat kernel keeps one dtype for both
                    # states; mixed requests stay fp32 (the safe side)
                    warnings.warn("precision-aware optimizer: exp_avg and "
                                  "exp_avg_sq dtypes differ; keeping both "
                                  "fp32")
            self.shard_m.append(torch.zeros_like(master, dtype=st_dt))
            self.shard_v.append(torch.zeros_like(master, dtype=st_dt))

            no_wd, norm_r = [], []
            for p, (s, e) in buf.param_index_map.items():
                os_, oe = max(s, lo), min(e, hi)
                if os_ >= oe:
                    continue
                r = (os_ - lo, oe - lo)
                if p.dim() == 1 or getattr(p, "_no_weight_decay", False):
                    no_wd.append(r)
                if param_is_not_tensor_parallel_duplicate(p):
                    norm_r.append(r)
            self.no_wd_ranges.append(no_wd)
            # coalesce adjacent ranges: at TP=1 every param counts, so
            # the whole shard collapses to ONE span and the norm/clip
            # pass is a single flat kernel instead of a foreach over
            # hundreds of per-param slices (1.7% of a MoE step)
            merged = []
            for r in sorted(norm_r):
                if merged and r[0] <= merged[-1][1]:
                    merged[-1] = (merged[-1][0], max(merged[-1][1], r[1]))
                else:
                    merged.append(r)
            self.norm_ranges.append(merged)

        # interface compat: param_groups for LR schedulers
        self.param_groups = [
            {"lr": self.lr, "wd_mult": 1.0, "lr_mult": 1.0, "is_decoupled_lr": False,
             "params": [], "weight_decay": config.weight_decay},
        ]

    # --------------------------------------------------------------
    def zero_grad(self, set_to_none: bool = True):
        for chunk in self.model_chunks:
            chunk.zero_grad_buffer()

    def get_loss_scale(self) -> torch.Tensor:
        device = "cuda" if torch.cuda.is_available() else "cpu"
        if self.grad_scaler is not None:
            return self.grad_scaler.scale.to(device)
        return torch.ones(1, dtype=torch.float32, device=device)

    def scale_loss(self, loss: torch.Tensor) -> torch.Tensor:
        """fp16: multiply the loss so small grads survive the fp16 range;
        wired into config.grad_scale_func by the training setup."""
        if self.grad_scaler is None:
            return loss
        return loss * self.grad_scaler.scale.to(loss.device)


    def _shard_grad(self, i):
        lo, hi = self.shard_bounds[i]
        return self.buffers[i].grad_data[lo:hi]

    def get_grad_norm(self) -> float:
        grads_for_norm = []
        for i, buf in enumerate(self.buffers):
            g = self._shard_grad(i)
            for (s, e) in self.norm_ranges[i]:
                grads_for_norm.append(g[s:e])
        extra = []
        if self.sharded:
            extra.append(parallel_state.get_data_parallel_group(
                with_context_parallel=True))
        return get_grad_norm_fp32(
            grads_for_norm,
            model_parallel_group=parallel_state.get_model_parallel_group(),
            extra_groups=extra)

    def count_zeros(self) -> float:
        grads = [self._shard_grad(i) for i in range(len(self.buffers))]
        return count_zeros_fp32(grads, parallel_state.get_model_parallel_group())

    def finish_param_sync(self):
        """Wait for an in-flight async param all-gather (overlap mode)."""
        for h in getattr(self, "_param_sync_handles", []):
            h.wait()
        self._param_sync_handles = []

    def _ranges_aligned(self, i) -> bool:
        """The single-launch ranged kernel decides wd per float4; it needs
        every no-wd boundary 4-aligned (true for transformer param sizes)."""
        return all(s % 4 == 0 and e % 4 == 0 for s, e in self.no_wd_ranges[i])

    def _nowd_device_ranges(self, i, device):
        """Merged, sorted no-wd ranges as cached device int64 tensors."""
        cache = getattr(self, "_nowd_cache", None)
        if cache is None:
            cache = self._nowd_cache = {}
        if i not in cache:
            merged = []
            for s, e in sorted(self.no_wd_ranges[i]):
                if merged and s <= merged[-1][1]:
                    merged[-1][1] = max(merged[-1][1], e)
                else:
                    merged.append([s, e])
            cache[i] = (
                torch.tensor([r[0] for r in merged], dtype=torch.int64,
                             device=device),
                torch.tensor([r[1] for r in merged], dtype=torch.int64,
                             device=device))
        return cache[i]

    @torch.no_grad()
    def step(self):
        self.step_count += 1
        from ..fp8 import bump_step
        bump_step()   # invalidate per-step fp8 weight-quantization caches
        lr = self.param_groups[0]["lr"]
        wd = self.param_groups[0].get("weight_decay", self.weight_decay)

        if self.grad_scaler is not None:
            inv = self.grad_scaler.inv_scale
            found_inf = False
            for i in range(len(self.buffers)):
                g = self._shard_grad(i)
                if not torch.isfinite(g).all():
                    found_inf = True
                g.mul_(inv)
            # Each rank only sees its own ZeRO shard (and its own TP/PP
            # slice of the model), so the skip decision must be agreed
            # globally or DynamicGradScaler state diverges per rank.
            if dist.is_initialized():
                dev = (torch.device("cuda", torch.cuda.current_device())
                       if torch.cuda.is_available() else torch.device("cpu"))
                flag = torch.tensor(
                    [1.0 if found_inf else 0.0], device=dev)
                dist.all_reduce(
                    flag, op=dist.ReduceOp.MAX,
                    group=parallel_state.get_model_parallel_group())
                dist.all_reduce(
                    flag, op=dist.ReduceOp.MAX,
                    group=parallel_state.get_data_parallel_group(
                        with_context_parallel=True))
                found_inf = bool(flag.item())
            self.grad_scaler.update(found_inf)
            if found_inf:
                for i in range(len(self.buffers)):
                    self._shard_grad(i).zero_()
                return False, None, None

        grad_norm = None
        if self.config.clip_grad > 0:
            grad_norm = self.get_grad_norm()
            if not math.isfinite(grad_norm):
                # skip update on inf/nan grad norm
                return False, grad_norm, None
            all_grads = [self._shard_grad(i) for i in range(len(self.buffers))]
            clip_grad_by_total_norm_fp32(all_grads, self.config.clip_grad,
                                         grad_norm)
        num_zeros = self.count_zeros() if self.config.log_num_zeros_in_grad else None

        with trace_scope("optimizer"):
            for i, buf in enumerate(self.buffers):
                grad = self._shard_grad(i)
                master, m, v = self.shard_master[i], self.shard_m[i], self.shard_v[i]
                no_wd = self.no_wd_ranges[i]
                lo, hi = self.shard_bounds[i]
                # bf16 param shard written by the adam kernel itself
                p16 = None
                if (buf.param_data is not None and master.is_cuda
                        and buf.param_data.dtype == torch.bfloat16):
                    p16 = buf.param_data[lo:hi]
                wrote_p16 = False
                if wd == 0 or not no_wd:
                    fused = _adam_step_flat(master, grad, m, v, lr,
                                            self.config.adam_beta1,
                                            self.config.adam_beta2,
                                            self.config.adam_eps, wd,
                                            self.step_count, p16=p16)
                    wrote_p16 = fused and p16 is not None
                elif (master.is_cuda and _ops.have_ops()
                      and hasattr(_ops.get_ops(), "adamw_flat_ranged")
                      and self._ranges_aligned(i)):
                    nw_s, nw_e = self._nowd_device_ranges(i, master.device)
                    _ops.get_ops().adamw_flat_ranged(
                        master, grad, m, v, nw_s, nw_e, lr,
                        self.config.adam_beta1, self.config.adam_beta2,
                        self.config.adam_eps, wd, self.step_count, p16)
                    wrote_p16 = p16 is not None
                else:
                    # two-pass: run with wd over the whole shard is wrong for
                    # no-wd params, so stitch: wd pass on full shard minus
                    # no-wd ranges is complex — instead run per-range.
                    cursor = 0
                    events = sorted(no_wd)
                    for (s, e) in events + [(master.numel(), master.numel())]:
                        if cursor < s:
                            _adam_step_flat(master[cursor:s], grad[cursor:s],
                                            m[cursor:s], v[cursor:s], lr,
                                            self.config.adam_beta1,
                                            self.config.adam_beta2,
                                            self.config.adam_eps, wd,
                                            self.step_count)
                        if s < e:
                            _adam_step_flat(master[s:e], grad[s:e], m[s:e],
                                            v[s:e], lr, self.config.adam_beta1,
                                            self.config.adam_beta2,
                                            self.config.adam_eps, 0.0,
                                            self.step_count)
                        cursor = max(cursor, e)

                # cast master back into model params (skipped when the
                # adam kernel already wrote the bf16 shard)
                if wrote_p16:
                    pass
                elif buf.param_data is not None:
                    buf.param_data[lo:hi].copy_(master)
                else:
                    for p, (s, e) in buf.param_index_map.items():
                        os_, oe = max(s, lo), min(e, hi)
                        if os_ < oe:
                            p.data.reshape(-1)[os_ - s:oe - s].copy_(
                                master[os_ - lo:oe - lo])

            # ZeRO-1: all-gather updated params; with overlap_param_gather
            # the gather runs async and the DDP forward pre-hook waits
            if self.sharded:
                overlap = getattr(self.config, "overlap_param_gather", False)
                self._param_sync_handles = []
                for buf in self.buffers:
                    if buf.ddp_config.use_distributed_optimizer:
                        h = buf.start_param_sync(async_op=overlap)
                        if h is not None:
                            self._param_sync_handles.append(h)

        return True, grad_norm, num_zeros

    # --- checkpointing ---------------------------------------------
    def state_dict(self):
        return {
            "step": self.step_count,
            "lr": self.param_groups[0]["lr"],
            "shard_master": [t.cpu() for t in self.shard_master],
            "shard_m": [t.cpu() for t in self.shard_m],
            "shard_v": [t.cpu() for t in self.shard_v],
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        self.param_groups[0]["lr"] = sd["lr"]
        for dst, src in zip(self.shard_master, sd["shard_master"]):
            dst.copy_(src.to(dst.device))
        for dst, src in zip(self.shard_m, sd["shard_m"]):
            dst.copy_(src.to(dst.device))
        for dst, src in zip(self.shard_v, sd["shard_v"]):
            dst.copy_(src.to(dst.device))

    def reload_model_params(self):
        for i, buf in enumerate(self.buffers):
            lo, hi = self.shard_bounds[i]
            if buf.param_data is not None:
                self.shard_master[i].copy_(buf.param_data[lo:hi].float())
            else:
                for p, (s, e) in buf.param_index_map.items():
                    os_, oe = max(s, lo), min(e, hi)
                    if os_ < oe:
                        self.shard_master[i][os_ - lo:oe - lo].copy_(
                            p.data.reshape(-1)[os_ - s:oe - s].float())
