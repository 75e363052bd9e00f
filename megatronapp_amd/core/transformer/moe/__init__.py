from .moe_layer import MoELayer, MoESubmodules
from .router import TopKRouter
