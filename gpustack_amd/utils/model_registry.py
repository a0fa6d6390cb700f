"""Architecture → model-category registry (reference:
scheduler/model_registry.py + meta_registry.py — the mapping that drives
`Model.categories`, placement sizing and the /v1/models metadata).

Text-serving framework: the supported categories are llm / embedding /
reranker (+ "moe" as an informational tag). Image/STT/TTS architectures
are recognized but map to their category so the control plane can reject
them with a clear 501 instead of a failed deployment (the OpenAI surface
returns structured `unsupported_modality` for those endpoints)."""
from __future__ import annotations

import json
from pathlib import Path

# exact-match table first, then prefix rules
_EXACT = {
    "BertForSequenceClassification": ["reranker"],
    "XLMRobertaForSequenceClassification": ["reranker"],
    "RobertaForSequenceClassification": ["reranker"],
    "BertModel": ["embedding"],
    "XLMRobertaModel": ["embedding"],
    "RobertaModel": ["embedding"],
}

_MOE_PREFIXES = ("Qwen3Moe", "Mixtral", "Glm4Moe", "Deepseek", "GptOss")
_UNSUPPORTED = {
    "WhisperForConditionalGeneration": ["speech_to_text"],
    "VitsModel": ["text_to_speech"],
    "StableDiffusionPipeline": ["image"],
    "FluxPipeline": ["image"],
}


def categories_for_architecture(arch: str) -> list[str]:
    if arch in _EXACT:
        return list(_EXACT[arch])
    if arch in _UNSUPPORTED:
        return list(_UNSUPPORTED[arch])
    if arch.endswith("ForSequenceClassification"):
        return ["reranker"]
    cats = ["llm"]
    if any(arch.startswith(p) for p in _MOE_PREFIXES):
        cats.append("moe")
    if arch.endswith(("Model", "ForMaskedLM")) and "CausalLM" not in arch:
        return ["embedding"]
    return cats


def categories_for_model(source: str, ref: str) -> list[str] | None:
    """Best-effort category derivation for a model record; None when the
    architecture cannot be determined (remote refs before download)."""
    arch = None
    if source == "preset":
        from ..engine.config import PRESETS

        spec = PRESETS.get(ref)
        arch = spec.architecture if spec else None
    elif str(ref).endswith(".gguf"):
        arch = "LlamaForCausalLM"  # GGUF serves through the llm path
    else:
        cfg = Path(ref) / "config.json"
        if cfg.exists():
            try:
                with open(cfg) as f:
                    arch = (json.load(f).get("architectures") or [None])[0]
            except (OSError, json.JSONDecodeError):
                arch = None
    return categories_for_architecture(arch) if arch else None
