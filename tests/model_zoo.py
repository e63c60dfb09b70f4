"""Back-compat shim: the canonical model zoo now lives in `npf.zoo`."""

from npf.zoo import *  # noqa: F401,F403
from npf.zoo import BUILDERS, PUBLISHED_PARAM_COUNTS, R_DIM, CNN_KWARGS  # noqa: F401
