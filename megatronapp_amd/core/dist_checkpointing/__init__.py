from .mapping import ShardedTensor, module_sharded_state_dict
from .serialization import load, load_common, save
