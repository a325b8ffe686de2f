from automodel_amd.quantization.nf4 import (  # noqa: F401
    NF4Linear,
    dequantize_nf4,
    quantize_linear_modules,
    quantize_nf4,
)
