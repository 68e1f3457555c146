from .trainer import Trainer, build_training  # noqa: F401
from .checkpoint import (save_checkpoint, Resumer,  # noqa: F401
                         load_model_from_checkpoint)
from .inference import infer_sequence, build_metrics  # noqa: F401
from .streaming import StreamingESR  # noqa: F401
from .graph_runner import GraphedBPTTStep, flatten_grads  # noqa: F401
