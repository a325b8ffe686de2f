"""Model registry + builder: HF architecture name -> in-tree MI355X model.

Reference behavior: nemo_automodel/_transformers/registry.py:485 (maps HF
``architectures`` to custom model classes) and auto_model.py:380-643
(meta-device init -> weight load / random init).
"""

from __future__ import annotations

import json
import os
from typing import Any

import torch

_REGISTRY: dict[str, Any] = {}


def register_architecture(names: tuple[str, ...]):
    def deco(cls):
        for n in names:
            _REGISTRY[n] = cls
        return cls

    return deco


def get_model_class(architecture: str):
    _ensure_builtin()
    if architecture not in _REGISTRY:
        raise KeyError(
            f"architecture '{architecture}' not registered; known: {sorted(_REGISTRY)}"
        )
    return _REGISTRY[architecture]


def _ensure_builtin() -> None:
    if _REGISTRY:
        return
    from automodel_amd.models.llama.model import LlamaForCausalLM

    for name in LlamaForCausalLM.hf_architectures:
        _REGISTRY[name] = LlamaForCausalLM
    try:
        from automodel_amd.moe.model import MoEForCausalLM

        for name in MoEForCausalLM.hf_architectures:
            _REGISTRY[name] = MoEForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.deepseek_v3.model import DeepseekV3ForCausalLM
        from automodel_amd.models.llama4.model import Llama4ForCausalLM

        for name in Llama4ForCausalLM.hf_architectures:
            _REGISTRY[name] = Llama4ForCausalLM
        from automodel_amd.models.nemotron_h.model import NemotronHForCausalLM

        for name in NemotronHForCausalLM.hf_architectures:
            _REGISTRY[name] = NemotronHForCausalLM
        from automodel_amd.models.bamba.model import BambaForCausalLM

        for name in BambaForCausalLM.hf_architectures:
            _REGISTRY[name] = BambaForCausalLM
        from automodel_amd.models.granitemoe_hybrid.model import (
            GraniteMoeHybridForCausalLM,
        )

        for name in GraniteMoeHybridForCausalLM.hf_architectures:
            _REGISTRY[name] = GraniteMoeHybridForCausalLM
        from automodel_amd.models.falcon_h1.model import FalconH1ForCausalLM

        for name in FalconH1ForCausalLM.hf_architectures:
            _REGISTRY[name] = FalconH1ForCausalLM
        from automodel_amd.models.qwen3_next.model import Qwen3NextForCausalLM

        for name in Qwen3NextForCausalLM.hf_architectures:
            _REGISTRY[name] = Qwen3NextForCausalLM
        from automodel_amd.models.lfm2.model import (
            Lfm2ForCausalLM,
            Lfm2MoeForCausalLM,
        )

        for name in Lfm2ForCausalLM.hf_architectures:
            _REGISTRY[name] = Lfm2ForCausalLM
        for name in Lfm2MoeForCausalLM.hf_architectures:
            _REGISTRY[name] = Lfm2MoeForCausalLM
        from automodel_amd.models.jamba.model import JambaForCausalLM

        for name in JambaForCausalLM.hf_architectures:
            _REGISTRY[name] = JambaForCausalLM
        from automodel_amd.models.zamba2.model import Zamba2ForCausalLM

        for name in Zamba2ForCausalLM.hf_architectures:
            _REGISTRY[name] = Zamba2ForCausalLM
        from automodel_amd.models.mamba_lm.model import (
            FalconMambaForCausalLM,
            Mamba2ForCausalLM,
            MambaForCausalLM,
        )

        for cls2 in (MambaForCausalLM, Mamba2ForCausalLM, FalconMambaForCausalLM):
            for name in cls2.hf_architectures:
                _REGISTRY[name] = cls2
        from automodel_amd.models.recurrent_gemma.model import (
            RecurrentGemmaForCausalLM,
        )

        for name in RecurrentGemmaForCausalLM.hf_architectures:
            _REGISTRY[name] = RecurrentGemmaForCausalLM
        from automodel_amd.models.phimoe.model import PhimoeForCausalLM

        for name in PhimoeForCausalLM.hf_architectures:
            _REGISTRY[name] = PhimoeForCausalLM
        from automodel_amd.models.gpt_bigcode.model import GPTBigCodeForCausalLM

        for name in GPTBigCodeForCausalLM.hf_architectures:
            _REGISTRY[name] = GPTBigCodeForCausalLM
        from automodel_amd.models.qwen3_vl.model import (
            Qwen3VLForConditionalGeneration,
            Qwen3VLMoeForConditionalGeneration,
        )

        for name in Qwen3VLForConditionalGeneration.hf_architectures:
            _REGISTRY[name] = Qwen3VLForConditionalGeneration
        for name in Qwen3VLMoeForConditionalGeneration.hf_architectures:
            _REGISTRY[name] = Qwen3VLMoeForConditionalGeneration
        from automodel_amd.models.glm4v.model import Glm4vForConditionalGeneration

        for name in Glm4vForConditionalGeneration.hf_architectures:
            _REGISTRY[name] = Glm4vForConditionalGeneration
        from automodel_amd.models.glm4v_moe.model import (
            Glm4vMoeForConditionalGeneration,
        )

        for name in Glm4vMoeForConditionalGeneration.hf_architectures:
            _REGISTRY[name] = Glm4vMoeForConditionalGeneration
        from automodel_amd.models.qwen2_audio.model import (
            Qwen2AudioForConditionalGeneration,
        )

        for name in Qwen2AudioForConditionalGeneration.hf_architectures:
            _REGISTRY[name] = Qwen2AudioForConditionalGeneration
        from automodel_amd.models.qwen2_5_omni.model import (
            Qwen2_5OmniThinkerForConditionalGeneration,
        )

        for name in Qwen2_5OmniThinkerForConditionalGeneration.hf_architectures:
            _REGISTRY[name] = Qwen2_5OmniThinkerForConditionalGeneration
        from automodel_amd.models.qwen3_omni_moe.model import (
            Qwen3OmniMoeThinkerForConditionalGeneration,
        )

        for name in Qwen3OmniMoeThinkerForConditionalGeneration.hf_architectures:
            _REGISTRY[name] = Qwen3OmniMoeThinkerForConditionalGeneration
        from automodel_amd.models.llava_onevision.model import (
            LlavaOnevisionForConditionalGeneration,
        )

        for name in LlavaOnevisionForConditionalGeneration.hf_architectures:
            _REGISTRY[name] = LlavaOnevisionForConditionalGeneration

        for name in DeepseekV3ForCausalLM.hf_architectures:
            _REGISTRY[name] = DeepseekV3ForCausalLM
        from automodel_amd.models.deepseek_v32.model import (
            DeepseekV32ForCausalLM,
        )

        for name in DeepseekV32ForCausalLM.hf_architectures:
            _REGISTRY[name] = DeepseekV32ForCausalLM
        from automodel_amd.models.kimi_linear.model import KimiLinearForCausalLM

        for name in KimiLinearForCausalLM.hf_architectures:
            _REGISTRY[name] = KimiLinearForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.gemma.model import Gemma3ForCausalLM, GemmaForCausalLM
        from automodel_amd.models.gemma.vlm import Gemma3ForConditionalGeneration

        for name in GemmaForCausalLM.hf_architectures:
            _REGISTRY[name] = GemmaForCausalLM
        for name in Gemma3ForCausalLM.hf_architectures:
            _REGISTRY[name] = Gemma3ForCausalLM
        for name in Gemma3ForConditionalGeneration.hf_architectures:
            _REGISTRY[name] = Gemma3ForConditionalGeneration
    except ImportError:
        pass
    try:
        from automodel_amd.models.gpt_oss.model import GptOssForCausalLM

        for name in GptOssForCausalLM.hf_architectures:
            _REGISTRY[name] = GptOssForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.glm4_moe.model import Glm4MoeForCausalLM

        for name in Glm4MoeForCausalLM.hf_architectures:
            _REGISTRY[name] = Glm4MoeForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.nemotron.model import NemotronForCausalLM

        for name in NemotronForCausalLM.hf_architectures:
            _REGISTRY[name] = NemotronForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.qwen2_vl.model import (
            Qwen2_5_VLForConditionalGeneration,
            Qwen2VLForConditionalGeneration,
        )

        for name in Qwen2VLForConditionalGeneration.hf_architectures:
            _REGISTRY[name] = Qwen2VLForConditionalGeneration
        for name in Qwen2_5_VLForConditionalGeneration.hf_architectures:
            _REGISTRY[name] = Qwen2_5_VLForConditionalGeneration
    except ImportError:
        pass
    try:
        from automodel_amd.models.starcoder2.model import Starcoder2ForCausalLM

        for name in Starcoder2ForCausalLM.hf_architectures:
            _REGISTRY[name] = Starcoder2ForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.stablelm.model import StableLmForCausalLM

        for name in StableLmForCausalLM.hf_architectures:
            _REGISTRY[name] = StableLmForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.codegen.model import CodeGenForCausalLM

        for name in CodeGenForCausalLM.hf_architectures:
            _REGISTRY[name] = CodeGenForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.bitnet.model import BitNetForCausalLM

        for name in BitNetForCausalLM.hf_architectures:
            _REGISTRY[name] = BitNetForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.hunyuan.model import HunYuanMoEV1ForCausalLM

        for name in HunYuanMoEV1ForCausalLM.hf_architectures:
            _REGISTRY[name] = HunYuanMoEV1ForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.minimax.model import MiniMaxForCausalLM

        for name in MiniMaxForCausalLM.hf_architectures:
            _REGISTRY[name] = MiniMaxForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.dbrx.model import DbrxForCausalLM

        for name in DbrxForCausalLM.hf_architectures:
            _REGISTRY[name] = DbrxForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.persimmon.model import PersimmonForCausalLM

        for name in PersimmonForCausalLM.hf_architectures:
            _REGISTRY[name] = PersimmonForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.gptj.model import GPTJForCausalLM

        for name in GPTJForCausalLM.hf_architectures:
            _REGISTRY[name] = GPTJForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.opt.model import OPTForCausalLM

        for name in OPTForCausalLM.hf_architectures:
            _REGISTRY[name] = OPTForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.bloom.model import BloomForCausalLM

        for name in BloomForCausalLM.hf_architectures:
            _REGISTRY[name] = BloomForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.mpt.model import MptForCausalLM

        for name in MptForCausalLM.hf_architectures:
            _REGISTRY[name] = MptForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.olmo.model import Olmo3ForCausalLM, OlmoForCausalLM

        for name in OlmoForCausalLM.hf_architectures:
            _REGISTRY[name] = OlmoForCausalLM
        for name in Olmo3ForCausalLM.hf_architectures:
            _REGISTRY[name] = Olmo3ForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.gpt_neox.model import GPTNeoXForCausalLM

        for name in GPTNeoXForCausalLM.hf_architectures:
            _REGISTRY[name] = GPTNeoXForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.arcee.model import ArceeForCausalLM

        for name in ArceeForCausalLM.hf_architectures:
            _REGISTRY[name] = ArceeForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.glm.model import GlmForCausalLM

        for name in GlmForCausalLM.hf_architectures:
            _REGISTRY[name] = GlmForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.exaone4.model import Exaone4ForCausalLM

        for name in Exaone4ForCausalLM.hf_architectures:
            _REGISTRY[name] = Exaone4ForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.cohere.model import (
            Cohere2ForCausalLM,
            CohereForCausalLM,
        )

        for name in CohereForCausalLM.hf_architectures:
            _REGISTRY[name] = CohereForCausalLM
        for name in Cohere2ForCausalLM.hf_architectures:
            _REGISTRY[name] = Cohere2ForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.phi.model import PhiForCausalLM

        for name in PhiForCausalLM.hf_architectures:
            _REGISTRY[name] = PhiForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.falcon.model import FalconForCausalLM

        for name in FalconForCausalLM.hf_architectures:
            _REGISTRY[name] = FalconForCausalLM
    except ImportError:
        pass
    try:
        from automodel_amd.models.gpt2.model import GPT2LMHeadModel

        for name in GPT2LMHeadModel.hf_architectures:
            _REGISTRY[name] = GPT2LMHeadModel
    except ImportError:
        pass
    try:
        from automodel_amd.models.mistral3.model import Mistral3ForConditionalGeneration

        for name in Mistral3ForConditionalGeneration.hf_architectures:
            _REGISTRY[name] = Mistral3ForConditionalGeneration
    except ImportError:
        pass
    try:
        from automodel_amd.models.llava.model import LlavaForConditionalGeneration

        for name in LlavaForConditionalGeneration.hf_architectures:
            _REGISTRY[name] = LlavaForConditionalGeneration
    except ImportError:
        pass
    try:
        from automodel_amd.models.vlm.model import VLMForConditionalGeneration

        for name in VLMForConditionalGeneration.hf_architectures:
            _REGISTRY[name] = VLMForConditionalGeneration
    except ImportError:
        pass


def build_model(
    config: dict | None = None,
    pretrained_path: str | None = None,
    architecture: str | None = None,
    backend: dict | None = None,
    dtype: str = "bfloat16",
    device: str | None = None,
    meta_init: bool = True,
) -> torch.nn.Module:
    """Construct a model from an arch-config dict or a local HF checkpoint dir.

    ``pretrained_path``: directory with HF ``config.json`` (+ safetensors —
    loaded by the checkpointer after sharding, reference checkpointing.py:1228).
    Random init happens on ``device`` when no weights are loaded.
    """
    if pretrained_path:
        with open(os.path.join(pretrained_path, "config.json")) as f:
            hf_cfg = json.load(f)
        architecture = architecture or hf_cfg.get("architectures", ["LlamaForCausalLM"])[0]
        try:
            cls = get_model_class(architecture)
        except KeyError:
            # generic HF-transformers fallback: any architecture without an
            # in-tree implementation still trains (reference
            # auto_model.py:380-643)
            from automodel_amd.models.hf_fallback import build_hf_fallback

            return build_hf_fallback(pretrained_path=pretrained_path,
                                     architecture=architecture, dtype=dtype,
                                     device=device)
        model_cfg = cls.config_from_hf(hf_cfg) if hasattr(cls, "config_from_hf") else None
        if model_cfg is None:
            from automodel_amd.models.llama.model import LlamaConfig

            model_cfg = LlamaConfig.from_hf_config(hf_cfg)
    else:
        assert config is not None, "build_model needs config= or pretrained_path="
        try:
            cls = get_model_class(architecture or "LlamaForCausalLM")
        except KeyError:
            from automodel_amd.models.hf_fallback import build_hf_fallback

            return build_hf_fallback(config=dict(config), architecture=architecture,
                                     dtype=dtype, device=device)
        cfg_cls = getattr(cls, "config_class", None)
        if cfg_cls is None:
            from automodel_amd.models.llama.model import LlamaConfig

            cfg_cls = LlamaConfig
        model_cfg = cfg_cls(**dict(config))

    torch_dtype = getattr(torch, dtype) if isinstance(dtype, str) else dtype
    init_device = torch.device("meta") if meta_init else None
    if init_device is not None:
        with init_device:
            model = cls(model_cfg, backend=backend)
    else:
        model = cls(model_cfg, backend=backend)
        # nn.Linear self-initializes but raw torch.empty Parameters (stacked
        # MoE experts, gates) do not — run the model's init so a non-meta
        # build never carries uninitialized memory.
        if hasattr(model, "init_weights"):
            model.init_weights()
        if device is not None and str(device) != "meta":
            model = model.to(device)
    model = model.to(dtype=torch_dtype)
    return model
