from .ddp import BucketedDataParallel  # noqa: F401
from .sampler import RoundRobinTaskSampler  # noqa: F401
