from .aggregators import (
    CyclicAggregator,
    GeoMedianAggregator,
    KrumAggregator,
    MeanAggregator,
    VoteAggregator,
)
from .comm import Communicator
from .flat import FlatSpace
from .trainer import Trainer

__all__ = [
    "Communicator",
    "FlatSpace",
    "Trainer",
    "MeanAggregator",
    "VoteAggregator",
    "GeoMedianAggregator",
    "KrumAggregator",
    "CyclicAggregator",
]
