from .clip_vit import CLIPViTModel, get_num_image_embeddings
from .multimodal_projector import MultimodalProjector
from .vit_layer_specs import get_vit_layer_local_spec
from .tasks import (
    DinoPretrainModel,
    VitClassificationModel,
    VitInpaintingModel,
    VitMlpHead,
)
