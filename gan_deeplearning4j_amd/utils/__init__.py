from .seed import seed_everything
from .profiling import rocprof_cmd, StepTimer
from .imaging import save_image_grid

__all__ = ["seed_everything", "rocprof_cmd", "StepTimer", "save_image_grid"]
