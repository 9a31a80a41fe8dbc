from .aggregate import GPUSharePlugin  # noqa: F401
from .config import AgentPaths, GPUPluginConfig, PluginOptions  # noqa: F401
