from .registry import build_model, compute_loss, SAMPLE_TOKENS
from .resnet import resnet18_cifar, resnet50
from .bert import BertMLM, BertConfig
from .llama import LlamaLM, LlamaConfig, LLAMA3_8B, LLAMA3_TINY

__all__ = ["build_model", "compute_loss", "SAMPLE_TOKENS",
           "resnet18_cifar", "resnet50", "BertMLM", "BertConfig",
           "LlamaLM", "LlamaConfig", "LLAMA3_8B", "LLAMA3_TINY"]
