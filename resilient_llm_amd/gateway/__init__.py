from .http import HttpServer, Request, Response  # noqa: F401
from .app import GatewayApp  # noqa: F401
