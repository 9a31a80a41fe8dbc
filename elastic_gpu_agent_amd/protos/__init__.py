from . import deviceplugin, podresources, protowire  # noqa: F401
