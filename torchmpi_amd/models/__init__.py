from .resnet import ResNet, resnet50
from .llama import LlamaConfig, LlamaModel, llama3_8b, llama_debug

__all__ = ["ResNet", "resnet50", "LlamaConfig", "LlamaModel", "llama3_8b",
           "llama_debug"]
