from .resnet import resnet50, resnet18  # noqa: F401
from .bert import bert_large, bert_tiny  # noqa: F401
