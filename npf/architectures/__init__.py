"""Ops/architectures layer: MLPs, attention, set convolutions, CNNs.

Public surface matches /root/reference/npf/architectures/__init__.py:1-6.
"""

from .attention import *  # noqa: F401,F403
from .cnn import *  # noqa: F401,F403
from .encoders import *  # noqa: F401,F403
from .mlp import *  # noqa: F401,F403
from .selfattn import *  # noqa: F401,F403
from .setcnn import *  # noqa: F401,F403
