"""`jit_fuser` API-parity shim (reference megatron/core/jit.py).

The reference decorates elementwise chains (bias+gelu, bias+dropout,
rope math, ...) with `jit_fuser` = torch.jit.script / torch.compile so
NVFuser stitches them into one kernel at trace time.  Here every hot
elementwise chain is a dedicated CDNA4 HIP kernel under `ops/csrc/`
(see docs/kernels.md), chosen at module level with an eager fallback —
tracing-compiler fusion is not part of the MI355X design.  `jit_fuser`
is therefore an identity decorator: code written against the reference
API keeps working, and the functions it marks stay plain eager Python
(their fused paths live in `megatronapp_amd.ops`).
"""


def jit_fuser(fn):
    return fn
