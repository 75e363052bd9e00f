"""Fully sharded data parallelism — ZeRO-3 (reference
core/distributed/custom_fsdp/fully_sharded_data_parallel.py, rebuilt
MI355X-first).

Each shard *unit* (by default one TransformerLayer; everything else forms
a root unit) flattens its params into one buffer sharded across the DP
group.  Lifecycle per step:

  pre-forward   all-gather the unit's full flat buffer (one RCCL
                all-gather over xGMI per unit, not per param)
  post-forward  free the full buffer (params point at a 0-size view)
  pre-backward  re-gather (full params needed for grad computation)
  post-grad     per-param grads accumulate into a full fp32 grad buffer;
                when the unit's last param grad arrives, reduce-scatter it
                to the local shard and free full params + full grads

Optimizer state (fp32 master/m/v) lives only on the local shard:
:meth:`optimizer_step` runs the same fused ranged-AdamW used by the
ZeRO-1 path on 1/dp of the parameters, then the next forward's
all-gathers broadcast the updated shards.

Peak parameter memory is param_bytes/dp + the largest unit's full size,
so 288 GB of HBM3E holds models dp× larger than DDP can.
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from .. import parallel_state
from ..optimizer.distrib_optimizer import _adam_step_flat


class _ShardUnit:
    """One flattened, DP-sharded group of parameters."""

    def __init__(self, name: str, module: torch.nn.Module,
                 params: List[torch.nn.Parameter], group):
        self.name = name
        self.module = module
        self.params = params
        self.group = group
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)

        numel = sum(p.numel() for p in params)
        self.pad_numel = math.ceil(numel / self.world) * self.world
        self.shard_numel = self.pad_numel // self.world
        dtype = params[0].dtype
        device = params[0].device

        flat = torch.zeros(self.pad_numel, dtype=dtype, device=device)
        self.offsets = []
        off = 0
        for p in params:
            flat[off:off + p.numel()].copy_(p.data.reshape(-1))
            self.offsets.append((off, off + p.numel()))
            off += p.numel()
        self.shard = flat[self.rank * self.shard_numel:
                          (self.rank + 1) * self.shard_numel].clone()
        # fp32 shard-grad accumulator (reduce-scatter target)
        self.shard_grad = torch.zeros(self.shard_numel, dtype=torch.float32,
                                      device=device)
        self._param_idx = {id(p): i for i, p in enumerate(params)}
        self.full: Optional[torch.Tensor] = None
        self.full_grad: Optional[torch.Tensor] = None
        self._pending = 0
        del flat
        self._release_params()

    # -- param materialization --------------------------------------------
    def gather(self):
        if self.full is not None:
            return
        self.full = torch.empty(self.pad_numel, dtype=self.shard.dtype,
                                device=self.shard.device)
        if self.world > 1:
            dist.all_gather_into_tensor(self.full, self.shard,
                                        group=self.group)
        else:
            self.full.copy_(self.shard)
        for p, (s, e) in zip(self.params, self.offsets):
            p.data = self.full[s:e].view(p._orig_shape)

    def _release_params(self):
        for p in self.params:
            if not hasattr(p, "_orig_shape"):
                p._orig_shape = p.shape
            p.data = torch.empty(0, dtype=p.dtype, device=p.device)
        self.full = None

    def release(self):
        if self.full is not None:
            self._release_params()

    # -- gradients ----------------------------------------------------------
    def begin_backward(self):
        self.gather()
        if self.full_grad is None:
            self.full_grad = torch.zeros(self.pad_numel, dtype=torch.float32,
                                         device=self.shard.device)
        self._pending = sum(1 for p in self.params if p.requires_grad)

    def grad_ready(self, param):
        idx = self._param_idx[id(param)]
        s, e = self.offsets[idx]
        if param.grad is not None:
            self.full_grad[s:e].add_(param.grad.reshape(-1).float())
            param.grad = None
        self._pending -= 1
        if self._pending == 0:
            if self.world > 1:
                self.full_grad.div_(self.world)
                dist.reduce_scatter_tensor(self.shard_grad, self.full_grad,
                                           group=self.group)
            else:
                self.shard_grad.copy_(self.full_grad)
            self.full_grad = None
            self._release_params()


class FullyShardedDataParallel(torch.nn.Module):
    """ZeRO-3 wrapper with an integrated sharded AdamW."""

    def __init__(self, module: torch.nn.Module, process_group=None,
                 lr: float = 1e-4, weight_decay: float = 0.0,
                 adam_betas=(0.9, 0.999), adam_eps: float = 1e-8,
                 clip_grad: float = 0.0):
        super().__init__()
        self.module = module
        self.group = (process_group if process_group is not None
                      else parallel_state.get_data_parallel_group())
        self.lr = lr
        self.weight_decay = weight_decay
        self.adam_betas = adam_betas
        self.adam_eps = adam_eps
        self.clip_grad = clip_grad
        self.step_count = 0

        # unit per transformer layer; leftovers form the root unit
        self.units: List[_ShardUnit] = []
        self._param_to_unit: Dict[torch.nn.Parameter, _ShardUnit] = {}
        claimed = set()
        for name, sub in module.named_modules():
            if type(sub).__name__ == "TransformerLayer":
                params = [p for p in sub.parameters() if p.requires_grad]
                if not params:
                    continue
                unit = _ShardUnit(name, sub, params, self.group)
                self.units.append(unit)
                claimed.update(id(p) for p in params)
                self._register_unit_hooks(sub, unit)
        rest = [p for p in module.parameters()
                if p.requires_grad and id(p) not in claimed]
        if rest:
            self.root_unit = _ShardUnit("root", module, rest, self.group)
            self.units.append(self.root_unit)
        else:
            self.root_unit = None

        self._grad_hooks = []
        for unit in self.units:
            for p in unit.params:
                self._param_to_unit[p] = unit
                self._grad_hooks.append(p.register_post_accumulate_grad_hook(
                    self._make_grad_hook(unit)))

        # fp32 master + moments per shard
        self.masters = [u.shard.float().clone() for u in self.units]
        self.m = [torch.zeros_like(ms) for ms in self.masters]
        self.v = [torch.zeros_like(ms) for ms in self.masters]

    # -- hooks ---------------------------------------------------------------
    def _register_unit_hooks(self, sub, unit):
        def pre_fwd(module, args, kwargs=None):
            unit.gather()

        def post_fwd(module, args, output):
            if module.training and torch.is_grad_enabled():
                # keep released until backward re-gathers
                unit.release()
            return output

        def pre_bwd(module, grad_output):
            unit.begin_backward()

        sub.register_forward_pre_hook(pre_fwd)
        sub.register_forward_hook(post_fwd)
        sub.register_full_backward_pre_hook(pre_bwd)

    def _make_grad_hook(self, unit):
        def hook(param):
            unit.grad_ready(param)
        return hook

    # -- API -----------------------------------------------------------------
    def forward(self, *args, **kwargs):
        # root unit (embeddings, final norm, head) stays live all step
        if self.root_unit is not None:
            self.root_unit.gather()
            if self.training and torch.is_grad_enabled():
                self.root_unit.begin_backward()
        return self.module(*args, **kwargs)

    def zero_grad_shards(self):
        for u in self.units:
            u.shard_grad.zero_()

    def grad_norm(self) -> float:
        sq = sum(float(u.shard_grad.norm() ** 2) for u in self.units)
        t = torch.tensor([sq])
        if dist.get_world_size(self.group) > 1:
            dist.all_reduce(t, group=self.group)
        return float(t.sqrt())

    @torch.no_grad()
    def optimizer_step(self):
        self.step_count += 1
        if self.clip_grad > 0:
            norm = self.grad_norm()
            if not math.isfinite(norm):
                self.zero_grad_shards()
                return False, norm
            clip = self.clip_grad / (norm + 1e-6)
            if clip < 1.0:
                for u in self.units:
                    u.shard_grad.mul_(clip)
        else:
            norm = None
        for u, master, m, v in zip(self.units, self.masters, self.m, self.v):
            _adam_step_flat(master, u.shard_grad, m, v, self.lr,
                            self.adam_betas[0], self.adam_betas[1],
                            self.adam_eps, self.weight_decay, self.step_count)
            u.shard.copy_(master.to(u.shard.dtype))
            u.release()  # next forward re-gathers updated shards
        self.zero_grad_shards()
        return True, norm
