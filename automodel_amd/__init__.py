"""automodel_amd — an MI355X-native fine-tune / pretrain framework.

A from-scratch AMD CDNA4 (gfx950) implementation of the capabilities of
NVIDIA-NeMo/Automodel: YAML-driven recipes, HF model loading, SPMD
parallelism (FSDP2/TP/SP/PP/CP/EP) over RCCL/xGMI, and hand-written HIP
kernels (MFMA + LDS tiling) for the hot ops.
"""

__version__ = "0.1.0"
