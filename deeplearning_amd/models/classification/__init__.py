from . import (coatnet, convnext, efficientnet, googlenet, lenet,  # noqa: F401
               repvgg, resnest, resnet, senet, shufflenet, swin, transfg, vgg,
               vit)
