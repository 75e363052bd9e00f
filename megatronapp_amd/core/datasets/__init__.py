from .gpt_dataset import (
    BlendedMegatronDatasetBuilder,
    GPTDataset,
    GPTDatasetConfig,
    MockGPTDataset,
)
