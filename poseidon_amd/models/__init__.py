from .zoo import (MODEL_ZOO, build_net, lenet, cifar10_quick, alexnet,
                  googlenet, vgg16)

__all__ = ["MODEL_ZOO", "build_net", "lenet", "cifar10_quick", "alexnet",
           "googlenet", "vgg16"]
