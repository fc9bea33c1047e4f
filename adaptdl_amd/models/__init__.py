"""Model zoo for the acceptance workloads (reference: examples/)."""

from adaptdl_amd.models.resnet import (ResNet18, ResNet34, ResNet50,  # noqa
                                       ResNet50Cifar)
