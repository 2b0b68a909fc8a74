"""Type aliases (reference ``gcbfplus/utils/typing.py``): keeps the public
API vocabulary of the reference; all are torch Tensors here."""
from typing import Any, Dict

from torch import Tensor

Array = Tensor
Action = Tensor  # (..., n_agents, action_dim)
Reward = Tensor
Cost = Tensor
Done = Tensor
Info = Dict[str, Any]
State = Tensor  # (..., state_dim)
AgentState = Tensor
Pos = Tensor
Pos2d = Tensor
Pos3d = Tensor
EdgeIndex = Tensor
Params = Dict[str, Tensor]
PRNGKey = Any  # numpy Generator in this build
BoolScalar = Tensor
Radius = float
