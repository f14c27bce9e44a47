from .helpers import (
    communicate,
    create_process_group,
    flatten_tensors,
    group_by_dtype,
    is_power_of,
    make_logger,
    unflatten_tensors,
)
from .metering import Meter

__all__ = [
    "communicate",
    "create_process_group",
    "flatten_tensors",
    "group_by_dtype",
    "is_power_of",
    "make_logger",
    "unflatten_tensors",
    "Meter",
]
