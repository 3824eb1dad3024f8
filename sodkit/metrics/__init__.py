from .sod import MAEMeter, PRFMeter, cal_mae, cal_pr_maxf  # noqa: F401
