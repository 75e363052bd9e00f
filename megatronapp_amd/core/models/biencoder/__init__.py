from .biencoder_model import (
    AllgatherFromDataParallelRegion,
    BiEncoderModel,
    PretrainedBertEncoder,
    biencoder_model_provider,
)
