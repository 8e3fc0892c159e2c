from .resnet import resnet18_cifar, resnet50_imagenet, ResNet
from .transformer import TranslationTransformer
from .lstm_lm import LSTMLanguageModel
from .recommendation import RecommendationAutoencoder
from .cyclegan import GeneratorResNet, Discriminator
from .a3c import ActorCritic

__all__ = [
    "resnet18_cifar",
    "resnet50_imagenet",
    "ResNet",
    "TranslationTransformer",
    "LSTMLanguageModel",
    "RecommendationAutoencoder",
    "GeneratorResNet",
    "Discriminator",
    "ActorCritic",
]
