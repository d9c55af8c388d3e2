from .cosine_lr import CosineLRScheduler  # noqa: F401
from .plateau_lr import PlateauLRScheduler  # noqa: F401
from .scheduler import Scheduler  # noqa: F401
from .scheduler_factory import create_scheduler  # noqa: F401
from .step_lr import StepLRScheduler  # noqa: F401
from .tanh_lr import TanhLRScheduler  # noqa: F401
