"""MegaFBD — forward/backward disaggregation.

Reference: megatron/virtual_tensor_parallel_communication.py,
megatron/Controller.py, parallel_state.py:204-963, schedules.py:2208-2505
(SURVEY.md §2.2).

MI355X-native redesign: each logical pipeline rank
is split into a FORWARD instance and a BACKWARD instance on different
GPUs — a clean process-level split instead of the reference's
thread-emulated TP ranks.  Forward instances run gradient-free forwards
and ship each stage's saved input activation to their dual backward
instance; backward instances recompute the forward with autograd and run
the backward + optimizer.  Collectives stay deadlock-free by
construction (disjoint half-DP groups, deterministic per-rank op order);
the reference's readiness controller (bitvector + DFS p2p reachability)
is kept as an optional gate (fbd/controller.py) for irregular schedules.
"""

from .topology import initialize_model_parallel_fbd
