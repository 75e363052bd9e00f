from .quantize import (
    calibrate_activation_scales,
    dequantize_weight,
    export_int8_state_dict,
    quantization_error,
    quantize_model,
    quantize_weight_int8,
)
