from .synthetic import SyntheticSOD, create_loader  # noqa: F401
