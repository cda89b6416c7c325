from .base import (
    NeuralNetworkModule,
    dynamic_module_wrapper,
    static_module_wrapper,
)
from .gaussian import GaussianPolicyHead
from .nature_cnn import ActorCriticCNN, NatureCNN, mlp
from .resnet import ResNet

__all__ = [
    "NeuralNetworkModule",
    "static_module_wrapper",
    "dynamic_module_wrapper",
    "NatureCNN",
    "ActorCriticCNN",
    "mlp",
    "ResNet",
    "GaussianPolicyHead",
]
