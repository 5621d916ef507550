from .token_bucket import MinuteWindowLimiter, RateSnapshot  # noqa: F401
from .core import (  # noqa: F401
    Router, DeploymentState, NoDeploymentAvailable, RouterRateLimit,
)
