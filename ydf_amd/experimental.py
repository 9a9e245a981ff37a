"""ydf.experimental.* analogue: the deep (neural) learners namespace
(PYDF gates these on JAX; here they run on PyTorch-ROCm directly)."""
from ydf_amd.deep import (DeepModel, MultiLayerPerceptronLearner,
                          TabularTransformerLearner)

MultiLayerPerceptronModel = DeepModel
TabularTransformerModel = DeepModel
