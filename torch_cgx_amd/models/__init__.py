from .resnet import resnet50  # noqa: F401
from .bert import bert_large  # noqa: F401
