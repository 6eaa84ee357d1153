"""Model zoo: the reference's experiment model families re-expressed in
PyTorch (channels-last on GPU, bf16-autocast friendly).

- MLP 784-100-10: reference experiments/mnist.py:132 (`_inference([784, 100,
  10], ...)`).
- CNNet: reference experiments/cnnet.py:58-95 (2x (conv5x5 + maxpool3x3/2) +
  dense 384 + dense 192 + linear 10, on 32x32x3 inputs).
- ResNet v1 families: reference external/slim/nets/resnet_v1.py:281+
  registers resnet_v1_{18,50,101,152,200}; re-implemented here as standard
  torchvision-style ResNets (BasicBlock / Bottleneck) plus the CIFAR-style
  resnet20/32/44/56/110 used by the CIFAR-10 configs in BASELINE.md.
"""

from .mlp import MLP
from .cnnet import CNNet
from .resnet import (resnet18, resnet34, resnet50, resnet101, resnet152,
                     resnet200, cifar_resnet20, cifar_resnet32, cifar_resnet44,
                     cifar_resnet56, cifar_resnet110, RESNETS)
from .vgg import vgg11, vgg16, vgg19
from .mobilenet import mobilenet_v2
from .inception import inception_v1, inception_v3

# Inception v3's padding-free grid reductions need a real spatial extent
# (the canonical input is 299; >= 75 keeps every grid >= 1). The
# experiment registry skips dataset shapes below this.
inception_v3.min_input = 75

#: Full model-factory map (the reference's slim ``networks_map`` analog,
#: external/slim/nets/nets_factory.py:39-66).
NETWORKS = dict(RESNETS)
NETWORKS.update({
    "vgg11": vgg11, "vgg16": vgg16, "vgg19": vgg19,
    "mobilenet_v2": mobilenet_v2,
    "inception_v1": inception_v1, "inception_v3": inception_v3,
})

__all__ = [
    "MLP", "CNNet", "RESNETS", "NETWORKS",
    "resnet18", "resnet34", "resnet50", "resnet101", "resnet152", "resnet200",
    "cifar_resnet20", "cifar_resnet32", "cifar_resnet44", "cifar_resnet56",
    "cifar_resnet110", "vgg11", "vgg16", "vgg19", "mobilenet_v2",
    "inception_v1", "inception_v3",
]
