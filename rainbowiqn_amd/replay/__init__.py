from .cpu_replay import ReplayMemory  # noqa: F401
from .sumtree import SumTree  # noqa: F401
