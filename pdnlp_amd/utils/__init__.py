from .seed import set_seed  # noqa: F401
from .logging import rank0_print, get_logger  # noqa: F401
from .metrics import MetricsWriter, StepTimer  # noqa: F401
from .checkpoint import save_checkpoint, load_checkpoint, strip_module_prefix  # noqa: F401
