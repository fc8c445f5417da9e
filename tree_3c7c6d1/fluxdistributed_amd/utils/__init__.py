from .device import device_ctx, synchronize, to_device, resolve_devices  # noqa: F401
from .metrics import topkaccuracy, maxk  # noqa: F401
from .timers import StageTimers, Throughput  # noqa: F401
from .checkpoint import save_checkpoint, load_checkpoint  # noqa: F401
from .logging import MetricsLogger, get_logger  # noqa: F401
