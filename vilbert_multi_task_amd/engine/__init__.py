from .runner import GraphRunner  # noqa: F401
