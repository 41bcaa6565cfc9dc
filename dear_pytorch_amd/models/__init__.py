"""Model zoo for the benchmark matrix (reference models came from torchvision
+ local inceptionv4 + transformers BERT; all are self-contained here)."""
from . import resnet, vgg, densenet, inceptionv4 as _iv4, bert, mnist
from .bert import (BertConfig, BertForPreTraining, BertPretrainingCriterion,
                   bert_base, bert_large)
from .mnist import MnistNet

_REGISTRY = {
    "resnet18": resnet.resnet18,
    "resnet34": resnet.resnet34,
    "resnet50": resnet.resnet50,
    "resnet101": resnet.resnet101,
    "resnet152": resnet.resnet152,
    "vgg11": vgg.vgg11,
    "vgg13": vgg.vgg13,
    "vgg16": vgg.vgg16,
    "vgg19": vgg.vgg19,
    "densenet121": densenet.densenet121,
    "densenet169": densenet.densenet169,
    "densenet201": densenet.densenet201,
    "inceptionv4": _iv4.inceptionv4,
    "mnistnet": MnistNet,
}


def get_cnn(name: str, num_classes: int = 1000, fused_bn: bool = False):
    name = name.lower()
    if name not in _REGISTRY:
        raise KeyError(f"unknown model '{name}'; have {sorted(_REGISTRY)}")
    if name == "mnistnet":
        return _REGISTRY[name]()
    if name.startswith(("resnet", "densenet", "inception")):
        return _REGISTRY[name](num_classes=num_classes, fused_bn=fused_bn)
    return _REGISTRY[name](num_classes=num_classes)


def list_models():
    return sorted(_REGISTRY) + ["bert_base", "bert_large"]
