from gllm_amd.layers.quantization.fp8 import (  # noqa: F401
    FP8_MAX, block_quant_fp8, convert_model_to_fp8, dequant_block_fp8,
    per_token_group_quant_fp8)
