from .seed import seed_everything
from .profiling import StepTimer, cuda_sync

__all__ = ["seed_everything", "StepTimer", "cuda_sync"]
