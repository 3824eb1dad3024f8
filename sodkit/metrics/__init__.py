"""SOD metrics: MAE / F-measure with distributed meters."""
from .sod import MAEMeter, PRFMeter, cal_mae, cal_pr_maxf  # noqa: F401
