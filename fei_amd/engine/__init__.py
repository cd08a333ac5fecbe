from fei_amd.engine.config import MODEL_SPECS, ModelSpec, get_spec

__all__ = ["MODEL_SPECS", "ModelSpec", "get_spec"]
