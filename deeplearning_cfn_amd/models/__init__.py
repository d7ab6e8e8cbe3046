from .resnet import build_resnet, ResNet  # noqa: F401
from .bert import build_bert, BertConfig, BertForPretraining  # noqa: F401
from .registry import build_model, MODEL_FAMILIES  # noqa: F401
