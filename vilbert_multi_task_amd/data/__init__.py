from .synthetic import synthetic_batch  # noqa: F401
