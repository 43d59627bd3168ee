from lzy_amd.models.resnet import resnet50
from lzy_amd.models.transformer import TransformerLM

__all__ = ["resnet50", "TransformerLM"]
