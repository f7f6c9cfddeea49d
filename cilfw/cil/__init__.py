from .memory import RehearsalMemory, herding_select
from .checkpoint import save_task_checkpoint, load_task_checkpoint

__all__ = ["RehearsalMemory", "herding_select", "save_task_checkpoint",
           "load_task_checkpoint"]
