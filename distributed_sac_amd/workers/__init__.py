from .trainer import Trainer  # noqa: F401
from .rollout import VecRollout  # noqa: F401
from .param_server import ParamSnapshot  # noqa: F401
