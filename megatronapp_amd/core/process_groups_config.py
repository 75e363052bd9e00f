"""Process-group bundles (reference core/process_groups_config.py).

Dataclasses that carry the communicators a model or grad-reduction path
needs, so modules can take explicit groups instead of reaching for the
parallel_state globals.  ``default()`` snapshots the current globals.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

import torch.distributed as dist

from . import parallel_state


@dataclass
class ModelCommProcessGroups:
    tp: Optional[dist.ProcessGroup] = None
    pp: Optional[dist.ProcessGroup] = None
    cp: Optional[dist.ProcessGroup] = None
    ep: Optional[dist.ProcessGroup] = None
    embd: Optional[dist.ProcessGroup] = None

    @classmethod
    def default(cls):
        return cls(
            tp=parallel_state.get_tensor_model_parallel_group(),
            pp=parallel_state.get_pipeline_model_parallel_group(),
            cp=parallel_state.get_context_parallel_group(),
            ep=parallel_state.get_expert_model_parallel_group(),
            embd=parallel_state.get_embedding_group())


@dataclass
class GradCommProcessGroups:
    dp: Optional[dist.ProcessGroup] = None
    dp_cp: Optional[dist.ProcessGroup] = None
    expt_dp: Optional[dist.ProcessGroup] = None

    @classmethod
    def default(cls):
        return cls(
            dp=parallel_state.get_data_parallel_group(),
            dp_cp=parallel_state.get_data_parallel_group(
                with_context_parallel=True),
            expt_dp=parallel_state.get_expert_data_parallel_group())
