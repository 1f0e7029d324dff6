from . import metrics  # noqa: F401
from .metrics import REGISTRY, Reporter, set_perf, stage_timer  # noqa: F401
