from .pods import Pod  # noqa: F401
