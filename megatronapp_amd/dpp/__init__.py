"""MegaDPP — dynamic pipeline-parallel transport.

Reference: megatron/shm_tensor_new_rdma*.cpp + training.py:742-788 wiring
(SURVEY.md §2.1/§2.2).  A POSIX-shm tagged-mailbox channel between PP
neighbours with a Python-side send-ordering policy (depth-first /
breadth-first / greedy FIFO).  Enabled with --use-dpp; the schedules tag
every pipeline send/recv with (model chunk, microbatch) and the receiver
blocks on exactly the tensor it needs, so the sender is free to reorder.
"""

from .transport import (
    DPPTransport,
    build_dpp_extension,
    get_transport,
    initialize_dpp,
    shutdown_dpp,
)
