from .mlp import MLP, Dense  # noqa
from .gnn import GNN, GNNLayer  # noqa
