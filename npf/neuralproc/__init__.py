"""Model layer: the 8 NPF members (reference npf/neuralproc/__init__.py:1-5)."""

from .attnnp import *  # noqa: F401,F403
from .base import *  # noqa: F401,F403
from .convnp import *  # noqa: F401,F403
from .gridconvnp import *  # noqa: F401,F403
from .np import *  # noqa: F401,F403
