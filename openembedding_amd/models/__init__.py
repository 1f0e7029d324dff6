from .criteo import CRITEO_FIELD_VOCABS, N_DENSE, N_SPARSE, synthetic_batch
from .ctr import CIN, LR, MODELS, WDL, DeepFM, xDeepFM

__all__ = ["CRITEO_FIELD_VOCABS", "N_DENSE", "N_SPARSE", "synthetic_batch",
           "CIN", "LR", "MODELS", "WDL", "DeepFM", "xDeepFM"]
