from .graphs import GraphedTrainStep  # noqa: F401
