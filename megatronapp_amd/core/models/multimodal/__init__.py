from .llava_model import (
    LLaVAModel, IGNORE_INDEX, IMAGE_TOKEN, DEFAULT_IMAGE_TOKEN_INDEX,
)
