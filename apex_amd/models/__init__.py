from .resnet import ResNet, resnet50
from .transformer import BertModel, GPTModel, TransformerLMConfig

__all__ = ["ResNet", "resnet50", "BertModel", "GPTModel", "TransformerLMConfig"]
