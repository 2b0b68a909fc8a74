from .cbf import CBFNet
from .distribution import TanhTransformedDistribution
from .policy import DeterministicPolicyNet, TanhNormalPolicyNet
from .value import ValueNet

__all__ = [
    "CBFNet", "DeterministicPolicyNet", "TanhNormalPolicyNet",
    "TanhTransformedDistribution", "ValueNet",
]
