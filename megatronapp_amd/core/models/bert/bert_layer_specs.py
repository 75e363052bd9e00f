"""BERT layer specs (reference models/bert/bert_layer_specs.py):
the GPT local spec with padding-mask attention."""

from ...enums import AttnMaskType
from ..gpt.gpt_layer_specs import get_gpt_layer_local_spec


def get_bert_layer_local_spec(qk_layernorm: bool = False,
                              normalization: str = "LayerNorm"):
    return get_gpt_layer_local_spec(
        qk_layernorm=qk_layernorm, normalization=normalization,
        use_flash=False, attn_mask_type=AttnMaskType.padding)
