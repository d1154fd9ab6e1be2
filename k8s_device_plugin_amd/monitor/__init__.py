from .feedback import FeedbackLoop  # noqa: F401
from .pathmon import PathMonitor  # noqa: F401
from .region import RegionSnapshot, SharedRegion, region_layout  # noqa: F401
