from .trace import RequestTrace, get_metrics, log_json  # noqa: F401
