from .model import DetectionModel, DetectorConfig  # noqa: F401
from .checkpoint import load_detectron_checkpoint, save_checkpoint  # noqa: F401
from .extractor import DetectorFeatureProvider  # noqa: F401
