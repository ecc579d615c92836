"""Distributed helpers: RCCL over xGMI intra-node, gloo on CPU."""
from . import dist
from .elastic import ElasticComm
from .pipeline import DataParallelPipeline, FanOutPipeline, StagePipeline

__all__ = [
    "dist",
    "ElasticComm",
    "DataParallelPipeline",
    "FanOutPipeline",
    "StagePipeline",
]
