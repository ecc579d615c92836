from .aggregator import OutputAggregator, OutputAggregatorConfig

__all__ = ["OutputAggregator", "OutputAggregatorConfig"]
