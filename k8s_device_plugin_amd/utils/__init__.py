from . import codec, kubeclient, nodelock, pendingpod, types  # noqa: F401
