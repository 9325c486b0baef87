"""Model zoo (reference example/image-classification/symbols/): resolve a
network name -> builder, mirroring `import symbols.<net>; net.get_symbol()`
(reference common/fit.py / train_imagenet.py)."""
from __future__ import annotations

from . import alexnet, inception_bn, inception_v3, lenet, mlp, resnet, vgg

_REGISTRY = {
    "resnet": resnet.get_symbol,
    "resnet-v1": resnet.get_symbol,
    "mlp": mlp.get_symbol,
    "lenet": lenet.get_symbol,
    "alexnet": alexnet.get_symbol,
    "vgg": vgg.get_symbol,
    "inception-v3": inception_v3.get_symbol,
    "inception-bn": inception_bn.get_symbol,
    "inceptionv3": inception_v3.get_symbol,
}


def get_symbol(network: str, **kwargs):
    if network not in _REGISTRY:
        raise ValueError(f"unknown network {network}; have {sorted(_REGISTRY)}")
    return _REGISTRY[network](**kwargs)
