from lzy_amd.models.resnet import resnet50

__all__ = ["resnet50"]
