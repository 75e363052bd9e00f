from . import parallel_state
from .enums import ModelType
from .transformer_config import MLATransformerConfig, ModelParallelConfig, TransformerConfig
from . import dist_checkpointing
from . import tensor_parallel

# commonly imported singletons (reference megatron.core.__init__)
from .inference_params import InferenceParams
from .num_microbatches_calculator import get_num_microbatches
from .packed_seq_params import PackedSeqParams

# reference-API alias: `from megatron.core import mpu` is parallel_state
from . import parallel_state as mpu
