from .grad_scaler import GradScaler  # noqa: F401
from .cast import cast_model_to, amp_dtype_of  # noqa: F401
