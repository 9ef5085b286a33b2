"""Serving plane: TF-Serving-compatible Predict over gRPC (:8500) and REST
(:8501), SavedModel-layout checkpoints, dynamic batching, model-version hot
reload — the MI355X-native replacement for SURVEY.md S1-S4 / N11.
"""

from .wireproto import (  # noqa: F401
    TensorProto, PredictRequest, PredictResponse, ModelSpec,
    tensor_to_numpy, numpy_to_tensor,
)
from .savedmodel import export_saved_model, load_saved_model  # noqa: F401
