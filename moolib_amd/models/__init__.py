from .atari import AtariNet, create_model  # noqa: F401
from .cartpole import CartPoleNet  # noqa: F401
