"""Model zoo for the acceptance workloads (reference: examples/)."""

from adaptdl_amd.models.resnet import (ResNet18, ResNet34, ResNet50,  # noqa
                                       ResNet50Cifar)
from adaptdl_amd.models.transformer import TransformerLM  # noqa: F401
from adaptdl_amd.models.bert import (BertConfig, BertModel,  # noqa: F401
                                     BertForMaskedLM, BertForPreTraining)
from adaptdl_amd.models.ncf import NeuMF  # noqa: F401
from adaptdl_amd.models.dcgan import Generator, Discriminator  # noqa: F401
from adaptdl_amd.models.cifar import CIFAR_MODELS  # noqa: F401,E402
