"""Deep tabular learners in PyTorch-ROCm (PYDF ydf/deep/ analogue)."""
from ydf_amd.deep.core import (DeepModel, MultiLayerPerceptronLearner,
                               TabularTransformerLearner)

__all__ = ["DeepModel", "MultiLayerPerceptronLearner",
           "TabularTransformerLearner"]
