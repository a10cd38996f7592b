from .vilbert import ViLBertModel  # noqa: F401
from .heads import VILBertForVLTasks  # noqa: F401
