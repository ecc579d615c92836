from .new_value import (
    NewValueComboDetector,
    NewValueDetector,
    NewValueDetectorConfig,
)
from .random_detector import RandomDetector, RandomDetectorConfig
from .dummy_detector import DummyDetector, DummyDetectorConfig

__all__ = [
    "NewValueDetector",
    "NewValueComboDetector",
    "NewValueDetectorConfig",
    "RandomDetector",
    "RandomDetectorConfig",
    "DummyDetector",
    "DummyDetectorConfig",
]
from .transformer import TransformerDetector, TransformerDetectorConfig

__all__ += ["TransformerDetector", "TransformerDetectorConfig"]
from .embedding import EmbeddingDetector, EmbeddingDetectorConfig

__all__ += ["EmbeddingDetector", "EmbeddingDetectorConfig"]
from .fused_pipeline import FusedPipelineDetector, FusedPipelineDetectorConfig

__all__ += ["FusedPipelineDetector", "FusedPipelineDetectorConfig"]
from .sklearn_detector import (
    FrequencyDetector,
    FrequencyDetectorConfig,
    SklearnDetector,
    SklearnDetectorConfig,
)

__all__ += ["SklearnDetector", "SklearnDetectorConfig",
            "FrequencyDetector", "FrequencyDetectorConfig"]
