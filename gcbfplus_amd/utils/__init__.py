from .graph import AGENT, GOAL, OBS, GraphBatch  # noqa
