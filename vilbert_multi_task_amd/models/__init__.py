from .vilbert import ViLBertModel  # noqa: F401
from .heads import VILBertForVLTasks  # noqa: F401
from .pretraining import BertForMultiModalPreTraining, BaseBertForVLTasks  # noqa: F401
