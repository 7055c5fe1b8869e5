from .trainer import Trainer, build_training  # noqa: F401
from .accel import Accelerator  # noqa: F401
from .hf_trainer import TrainingArguments, HFStyleTrainer  # noqa: F401
from .fabric import Fabric  # noqa: F401
from .infer import InferenceEngine  # noqa: F401
