from .seed import seed_everything
from .profiling import rocprof_cmd, StepTimer

__all__ = ["seed_everything", "rocprof_cmd", "StepTimer"]
