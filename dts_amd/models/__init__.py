from dts_amd.models.config import MODEL_REGISTRY, ModelSpec, get_model_spec

__all__ = ["ModelSpec", "MODEL_REGISTRY", "get_model_spec"]
