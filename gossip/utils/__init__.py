from stochastic_gradient_push_amd.utils import (
    Meter,
    communicate,
    create_process_group,
    flatten_tensors,
    group_by_dtype,
    is_power_of,
    make_logger,
    unflatten_tensors,
)

__all__ = [
    "Meter", "communicate", "create_process_group", "flatten_tensors",
    "group_by_dtype", "is_power_of", "make_logger", "unflatten_tensors",
]
