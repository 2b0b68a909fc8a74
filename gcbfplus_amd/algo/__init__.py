"""Algorithm factory (reference gcbfplus/algo/__init__.py:8-18)."""
from __future__ import annotations

from .base import MultiAgentController
from .gcbf import GCBF
from .gcbf_plus import GCBFPlus

_ALGOS = {"gcbf": GCBF, "gcbf+": GCBFPlus}

try:
    from .centralized_cbf import CentralizedCBF

    _ALGOS["centralized_cbf"] = CentralizedCBF
except ImportError:
    pass
try:
    from .dec_share_cbf import DecShareCBF

    _ALGOS["dec_share_cbf"] = DecShareCBF
except ImportError:
    pass


def make_algo(algo: str, **kwargs) -> MultiAgentController:
    if algo not in _ALGOS:
        raise ValueError(f"Unknown algorithm: {algo}")
    return _ALGOS[algo](**kwargs)
