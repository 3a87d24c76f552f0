"""Policy infrastructure: network DSL parser, custom layers, functional
(flat-parameter, population-batched) modules, Policy.

Reference parity: /root/reference/src/evotorch/neuroevolution/net/.
"""

from .functional import ModuleExpectingFlatParameters, make_functional_module
from .layers import LSTM, RNN, Apply, Bin, Clip, FeedForwardNet, LocomotorNet, Round, Slice, StructuredControlNet
from .misc import count_parameters, device_of_module, fill_parameters, parameter_vector
from .multilayered import MultiLayered
from .parser import NetParsingError, str_to_net
from .policy import Policy, reset_tensors
from .statefulmodule import StatefulModule, ensure_stateful

__all__ = [
    "Apply",
    "Bin",
    "Clip",
    "FeedForwardNet",
    "LocomotorNet",
    "LSTM",
    "ModuleExpectingFlatParameters",
    "MultiLayered",
    "NetParsingError",
    "Policy",
    "RNN",
    "Round",
    "Slice",
    "StatefulModule",
    "StructuredControlNet",
    "count_parameters",
    "device_of_module",
    "ensure_stateful",
    "fill_parameters",
    "make_functional_module",
    "parameter_vector",
    "reset_tensors",
    "str_to_net",
]
