"""Model zoo — native implementations of the reference's full inventory
(reference models/__init__.py:16-30 plus the torchvision-backed entries in
dl_trainer.py:87-135; SURVEY.md §2.1 "Models package")."""
from .resnet_cifar import (resnet20, resnet32, resnet44, resnet56,
                           resnet110, resnet_mod20, resnet_mod32,
                           resnet_mod44, resnet_mod56, resnet_mod110,
                           CifarResNet)
from .resnet_imagenet import (resnet18, resnet34, resnet50, resnet101,
                              resnet152, ResNet)
from .preresnet import (preresnet20, preresnet32, preresnet44, preresnet56,
                        preresnet110)
from .resnext import resnext29_8_64, resnext29_16_64
from .densenet import (densenet100_12, densenet121, densenet161,
                       densenet201)
from .vgg import VGG, vgg16, vgg19, vgg16i
from .alexnet import AlexNet, alexnet
from .googlenet import googlenet, GoogLeNet
from .inception import inceptionv3, inceptionv4
from .caffe_cifar import caffe_cifar
from .small import LeNet, MnistNet, FCN5Net, LinearRegression
from .lstm import lstm, repackage_hidden
from .deepspeech import DeepSpeech, LSTMAN4, GreedyDecoder

__all__ = [
    'resnet20', 'resnet32', 'resnet44', 'resnet56', 'resnet110',
    'resnet_mod20', 'resnet_mod32', 'resnet_mod44', 'resnet_mod56',
    'resnet_mod110', 'resnet18', 'resnet34', 'resnet50', 'resnet101',
    'resnet152', 'preresnet20', 'preresnet32', 'preresnet44',
    'preresnet56', 'preresnet110', 'resnext29_8_64', 'resnext29_16_64',
    'densenet100_12', 'densenet121', 'densenet161', 'densenet201', 'VGG',
    'vgg16', 'vgg19', 'vgg16i', 'AlexNet', 'alexnet', 'googlenet',
    'inceptionv3', 'inceptionv4', 'caffe_cifar', 'LeNet', 'MnistNet',
    'FCN5Net', 'LinearRegression', 'lstm', 'repackage_hidden',
    'DeepSpeech', 'LSTMAN4', 'GreedyDecoder',
]
