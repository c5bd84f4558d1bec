from .plan import *  # noqa: F401,F403
from .builder import LogicalPlanBuilder  # noqa: F401
