from byzpy_amd.pre_aggregators.base import PreAggregator
from byzpy_amd.pre_aggregators.ops import ARC, Bucketing, Clipping, NearestNeighborMixing

NNM = NearestNeighborMixing

__all__ = ["PreAggregator", "Clipping", "Bucketing", "NearestNeighborMixing", "NNM", "ARC"]
