from . import (coatnet, convnext, efficientnet, googlenet, lenet,  # noqa: F401
               repvgg, resnest, resnet, senet, shufflenet, swin, swin_moe,
               transfg, vgg, zoo_extra, zoo_tail,
               vit)
