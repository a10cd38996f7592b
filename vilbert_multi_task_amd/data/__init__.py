from .synthetic import synthetic_batch  # noqa: F401
from .loaders import ConceptCapLoaderTrain, ConceptCapLoaderVal, LoadDatasetEval  # noqa: F401
