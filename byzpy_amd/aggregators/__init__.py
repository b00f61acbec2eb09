from byzpy_amd.aggregators.base import Aggregator
from byzpy_amd.aggregators.coordinate_wise import (
    CoordinateWiseMedian,
    CoordinateWiseTrimmedMean,
    MeanOfMedians,
)
from byzpy_amd.aggregators.geometric_wise import (
    GeometricMedian,
    Krum,
    MinimumDiameterAveraging,
    MoNNA,
    MultiKrum,
    SMEA,
)
from byzpy_amd.aggregators.norm_wise import (
    CAF,
    CenteredClipping,
    ComparativeGradientElimination,
)

__all__ = [
    "Aggregator",
    "CoordinateWiseMedian",
    "CoordinateWiseTrimmedMean",
    "MeanOfMedians",
    "MultiKrum",
    "Krum",
    "GeometricMedian",
    "MinimumDiameterAveraging",
    "MoNNA",
    "SMEA",
    "CenteredClipping",
    "ComparativeGradientElimination",
    "CAF",
]
