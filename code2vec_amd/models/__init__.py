from .model_base import (Code2VecModelBase, ModelEvaluationResults,  # noqa: F401
                         ModelPredictionResults)
from .torch_model import Code2VecModel  # noqa: F401
