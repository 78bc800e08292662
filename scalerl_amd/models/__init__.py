from .atari import AtariNet, AtariQNet
from .mlp import ActorCriticNet, ActorNet, CriticNet, QNet
from .a3c_atari import A3CAtariNet
from .resnet import ResNetLSTMPolicy
from .noisy import CategoricalQNet, NoisyLinear, c51_loss

__all__ = ["AtariNet", "AtariQNet", "QNet", "ActorNet", "CriticNet", "ActorCriticNet",
           "A3CAtariNet", "ResNetLSTMPolicy", "CategoricalQNet", "NoisyLinear", "c51_loss"]
