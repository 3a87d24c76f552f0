"""L1 tensor substrate: dtype/device utilities, tensor factories, ranking,
object containers, immutability, cloning, hooks, constraints.

MI355X-native re-design of the reference's `tools` namespace
(/root/reference/src/evotorch/tools/__init__.py).
"""

from . import cloning, constraints, immutable, misc, ranking
from .cloning import Clonable, ReadOnlyClonable, Serializable, clone, deep_clone
from .constraints import log_barrier, penalty, violation
from .hook import Hook
from .immutable import ImmutableContainer, ImmutableDict, ImmutableList, ImmutableSet, as_immutable, mutable_copy
from .misc import (
    Device,
    DType,
    ErroneousResult,
    RealOrVector,
    Size,
    Vector,
    cast_tensors_in_container,
    clip_tensor,
    device_of,
    device_of_container,
    dtype_of,
    dtype_of_container,
    empty_tensor_like,
    ensure_tensor_length_and_dtype,
    expect_none,
    is_bool,
    is_bool_vector,
    is_dtype_bool,
    is_dtype_float,
    is_dtype_integer,
    is_dtype_object,
    is_dtype_real,
    is_integer,
    is_integer_vector,
    is_real,
    is_real_vector,
    is_sequence,
    make_empty,
    make_gaussian,
    make_gaussian_shaped_like,
    make_I,
    make_nan,
    make_ones,
    make_randint,
    make_tensor,
    make_uniform,
    make_uniform_shaped_like,
    make_zeros,
    modify_tensor,
    modify_vector,
    numpy_copy,
    pass_info_if_needed,
    as_tensor,
    is_tensor_on_cpu,
    multiply_rows_by_scalars,
    rowwise_sum,
    message_from,
    set_default_logger_config,
    make_batched_false_for_vmap,
    split_workload,
    stdev_from_radius,
    storage_ptr,
    to_stdev_init,
    to_numpy_dtype,
    to_torch_dtype,
)
from .objectarray import ObjectArray, as_object_array
from .structures import CBag, CDict, CList, CMemory, Structure, do_where
from .tensorframe import TensorFrame
from .ranking import centered, linear, nes, normalized, rank, raw
from .readonlytensor import ReadOnlyTensor, as_read_only_tensor, read_only_tensor
from .recursiveprintable import RecursivePrintable
from .tensormaker import TensorMakerMixin

__all__ = [n for n in dir() if not n.startswith("_")]
