from .chebconv import ChebConvStack  # noqa: F401
