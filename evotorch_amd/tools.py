"""`evotorch_amd.tools` — alias of `evotorch_amd.utils` for reference-API
familiarity (the reference exposes this namespace as `evotorch.tools`)."""

from .utils import *  # noqa: F401,F403
from .utils import __all__  # noqa: F401
