from .mlp import FusedMLP, build_mlp, weights_init  # noqa: F401
from .actor import Actor, LLActor  # noqa: F401
from .critic import Critic, LLCritic  # noqa: F401
