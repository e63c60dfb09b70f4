"""npf: MI355X-native Neural Process Family framework.

Public API parity with the reference package root
(/root/reference/npf/__init__.py:1-2): models + losses at top level, plus
`npf.architectures`, `npf.utils.datasplit`, and the MI355X additions
`npf.ops` (fused HIP kernels), `npf.data`, `npf.train`, `npf.parallel`.
"""

__version__ = "0.1.0"

import torch.distributions as _dists

# torch.distributions argument validation calls `.all()` on construction:
# a host<->device sync on EVERY forward (the predictive distribution is built
# each step) and illegal inside hipGraph capture.  The math is unchanged;
# invalid parameters surface as NaNs instead of eager ValueErrors.
_dists.Distribution.set_default_validate_args(False)

from . import ops  # noqa: F401  (import first: architectures depend on it)
from .losses import *  # noqa: F401,F403
from .neuralproc import *  # noqa: F401,F403
