from .logging import log_with_timestamp, sanitize_error  # noqa: F401
