from .config import PluginConfig, parse_args  # noqa: F401
from .kfd import PhysicalGPU, enumerate_gpus, kfd_healthy  # noqa: F401
from .rm import ResourceManager  # noqa: F401
from .server import VGPUDevicePlugin  # noqa: F401
