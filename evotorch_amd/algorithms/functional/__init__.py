"""Functional (ask/tell, pure-function) searchers and optimizers.

Reference parity: /root/reference/src/evotorch/algorithms/functional/
(funcpgpe.py:67,301,330; funccem.py:34,235,249; funcadam.py:34;
funcclipup.py:31; funcsgd.py:30; misc.py:26).

Each algorithm is an `alg(...) -> State` / `alg_ask(state, ...)` /
`alg_tell(state, values, evals) -> State` triple whose state is a
NamedTuple of tensors. All operations are plain batched tensor math over
the trailing dimensions, so adding leading batch dimensions to the state
runs B independent searches — no object state, vmap-free batching.
"""

from .funccem import CEMState, cem, cem_ask, cem_tell
from .funcoptimizers import (
    AdamState,
    OptimizerFunctions,
    get_functional_optimizer,
    ClipUpState,
    SGDState,
    adam,
    adam_ask,
    adam_tell,
    clipup,
    clipup_ask,
    clipup_tell,
    sgd,
    sgd_ask,
    sgd_tell,
)
from .funcpgpe import PGPEState, pgpe, pgpe_ask, pgpe_tell
from .funcsnes import SNESState, snes, snes_ask, snes_tell

__all__ = [
    "AdamState",
    "CEMState",
    "ClipUpState",
    "PGPEState",
    "SGDState",
    "adam",
    "adam_ask",
    "adam_tell",
    "cem",
    "cem_ask",
    "cem_tell",
    "clipup",
    "clipup_ask",
    "clipup_tell",
    "pgpe",
    "pgpe_ask",
    "pgpe_tell",
    "sgd",
    "sgd_ask",
    "sgd_tell",
]
