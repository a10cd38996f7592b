"""MI355X-native ViLBERT 12-in-1 serving + multi-task training framework.

From-scratch rebuild of the capabilities of Cloud-CV/vilbert-multi-task
(reference at /root/reference, analyzed in SURVEY.md) designed gfx950-first:
hand-written CDNA4 HIP kernels for the transformer hot ops, RCCL over xGMI
for data-parallel training, hipGraph-captured batched forwards for serving.
"""

__version__ = "0.1.0"

from .config import ViLBertConfig  # noqa: F401
from .tasks import TASKS, TaskSpec, get_task  # noqa: F401
