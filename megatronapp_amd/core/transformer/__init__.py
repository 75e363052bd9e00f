from ..transformer_config import TransformerConfig
from .module import MegatronModule
from .spec_utils import ModuleSpec, build_module
from .transformer_block import TransformerBlock
from .transformer_layer import TransformerLayer, TransformerLayerSubmodules
