from .model import DetectionModel, DetectorConfig  # noqa: F401
from .extractor import DetectorFeatureProvider  # noqa: F401
