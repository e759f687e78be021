from .model import NoisyLinear, RainbowIQN  # noqa: F401
