# Copyright (c) Flashy-AMD authors.
"""Model zoo for the example workloads and benchmarks.

The reference has no model zoo (it borrows torchvision's resnet18,
/root/reference/examples/cifar/train.py:43); torchvision is not a dependency
here, so the architectures the benchmark configs name are implemented
in-house: ResNet-18/50 (CIFAR and ImageNet stems) and the DCGAN-style 64x64
generator/discriminator pair used by the adversarial workload.
"""
from .resnet import ResNet, resnet18, resnet50  # noqa: F401
from .resnet_native import (NativeResNet, native_resnet18,  # noqa: F401
                            native_resnet50)
from .dcgan import DCGANGenerator, DCGANDiscriminator  # noqa: F401
from .dcgan_native import (NativeDCGANGenerator,  # noqa: F401
                           NativeDCGANDiscriminator)
