from fei_amd.models.llama import LlamaModel

__all__ = ["LlamaModel"]
