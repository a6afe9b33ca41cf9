from .optimizer import GalvatronOptimizer, get_optimizer_and_param_scheduler  # noqa: F401
from .scheduler import OptimizerParamScheduler  # noqa: F401
