from .arena import ParamArena, ParamSpec  # noqa: F401
from .vgg import TaskBatchedVGG  # noqa: F401
