"""Per-module kernel backend selection.

Reference behavior: nemo_automodel/components/models/common/utils.py:282-370
(BackendConfig selects attn/linear/rms_norm/rope/experts implementations per
model). On MI355X the choices are: hand-written HIP/CDNA4 kernels ("hip"),
plain torch eager ("torch"), or torch SDPA for attention ("sdpa").
"""

from __future__ import annotations

from dataclasses import dataclass, replace


@dataclass
class BackendConfig:
    # attention: "hip" = in-tree CDNA4 flash kernel, "sdpa" = torch SDPA,
    # "eager" = explicit matmul+softmax reference.
    attn: str = "hip"
    rms_norm: str = "hip"
    rope: str = "hip"
    # cross-entropy: "hip_fused" = fused lm_head-GEMM+CE without materializing
    # logits (cut-cross-entropy equivalent), "chunked" = chunked logits CE,
    # "torch" = plain CE.
    loss: str = "hip_fused"
    linear: str = "torch"          # plain GEMMs ride hipBLASLt via torch.linear
    experts: str = "hip_grouped"   # MoE expert compute
    dispatcher: str = "rccl_a2a"   # MoE token dispatch

    def for_cpu(self) -> "BackendConfig":
        """CPU-safe variant (unit tests run without a GPU)."""
        return BackendConfig(
            attn="sdpa",
            rms_norm="torch",
            rope="torch",
            loss="chunked",
            linear="torch",
            experts="torch",
            dispatcher="torch",
        )

    # head dims the in-tree flash kernels are tiled for (flash_attn.hip);
    # any dim <= FLASH_MAX_HEAD_DIM runs on the kernels (the wrapper
    # zero-pads to the next tile size — ops/attention.py _target_dims).
    FLASH_HEAD_DIMS = (64, 96, 128, 192, 256)
    FLASH_MAX_HEAD_DIM = 256

    @classmethod
    def resolve(cls, cfg: "BackendConfig | dict | None", device_type: str,
                head_dim: int | None = None) -> "BackendConfig":
        if cfg is None:
            cfg = cls()
        elif isinstance(cfg, dict):
            cfg = cls(**cfg)
        if device_type != "cuda":
            return cfg.for_cpu()
        if (head_dim is not None and cfg.attn == "hip"
                and head_dim > cls.FLASH_MAX_HEAD_DIM):
            # explicit, visible downgrade at model build — NOT a silent
            # runtime fallback (the kernel itself still fails loudly)
            import warnings

            warnings.warn(
                f"head_dim {head_dim} > flash kernel max "
                f"{cls.FLASH_MAX_HEAD_DIM}; using sdpa attention for this model",
                stacklevel=2,
            )
            cfg = replace(cfg, attn="sdpa")
        return cfg
