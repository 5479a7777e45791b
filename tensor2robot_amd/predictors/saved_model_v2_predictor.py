"""Servable-loading predictor aliases (reference saved_model_v2_predictor).

The reference ships three SavedModel predictor flavors (base/TF1/TF2,
`predictors/saved_model_v2_predictor.py:33/211/231`); in the torch-native
framework there is one servable format (TorchScript + t2r_assets), so
all three names resolve to ExportedSavedModelPredictor.
"""

from tensor2robot_amd.predictors.exported_savedmodel_predictor import (
    ExportedSavedModelPredictor,
)

SavedModelPredictorBase = ExportedSavedModelPredictor
SavedModelTF1Predictor = ExportedSavedModelPredictor
SavedModelTF2Predictor = ExportedSavedModelPredictor

__all__ = ["SavedModelPredictorBase", "SavedModelTF1Predictor",
           "SavedModelTF2Predictor", "ExportedSavedModelPredictor"]
