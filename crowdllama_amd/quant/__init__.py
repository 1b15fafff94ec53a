from .kquants import (  # noqa: F401
    QK_K, Q4_K_BLOCK_BYTES, Q6_K_BLOCK_BYTES, Q8_0_BLOCK, Q8_0_BLOCK_BYTES,
    quantize_q4_k, dequantize_q4_k,
    quantize_q6_k, dequantize_q6_k,
    quantize_q8_0, dequantize_q8_0,
    quantize, dequantize, GGMLType, type_block_bytes, row_bytes,
)
from .gguf import GGUFReader, GGUFWriter, GGUFTensorInfo  # noqa: F401
