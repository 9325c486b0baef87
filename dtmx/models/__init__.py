"""Model zoo (reference example/image-classification/symbols/): resolve a
network name -> builder, mirroring `import symbols.<net>; net.get_symbol()`
(reference common/fit.py / train_imagenet.py)."""
from __future__ import annotations

from . import (alexnet, googlenet, inception_bn, inception_resnet_v2,
               inception_v3, inception_v4, lenet, mlp, mobilenet, resnet,
               resnet_v2, resnext, vgg)

_REGISTRY = {
    "resnet": resnet.get_symbol,
    # dtmx's ResNet blocks are post-activation with the fused
    # residual+relu BN epilogue — the v1 topology (reference
    # resnet-v1.py); the name "resnet" serves both spellings
    "resnet-v1": resnet.get_symbol,
    "mlp": mlp.get_symbol,
    "lenet": lenet.get_symbol,
    "alexnet": alexnet.get_symbol,
    "vgg": vgg.get_symbol,
    "googlenet": googlenet.get_symbol,
    "inception-v3": inception_v3.get_symbol,
    "inception-bn": inception_bn.get_symbol,
    "inceptionv3": inception_v3.get_symbol,
    "inception-v4": inception_v4.get_symbol,
    "inception-resnet-v2": inception_resnet_v2.get_symbol,
    "mobilenet": mobilenet.get_symbol,
    "mobilenetv2": mobilenet.get_symbol_v2,
    "resnext": resnext.get_symbol,
    # the reference's own resnet.py topology (pre-activation v2)
    "resnet-v2": resnet_v2.get_symbol,
}


def get_symbol(network: str, **kwargs):
    if network not in _REGISTRY:
        raise ValueError(f"unknown network {network}; have {sorted(_REGISTRY)}")
    return _REGISTRY[network](**kwargs)
