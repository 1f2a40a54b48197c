from .resnet import ResNet, resnet50
from .transformer import (
    BertModel,
    GPTModel,
    LlamaModel,
    TransformerLargeModel,
    TransformerLMConfig,
)

__all__ = ["ResNet", "resnet50", "BertModel", "GPTModel", "LlamaModel",
           "TransformerLargeModel", "TransformerLMConfig"]
