from automodel_amd.serving.decode_linear import (  # noqa: F401
    DecodeLinear,
    gemv_bf16,
    swap_linears_for_decode,
)
