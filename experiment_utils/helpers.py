from stochastic_gradient_push_amd.utils.helpers import make_logger
from stochastic_gradient_push_amd.utils.nic import get_tcp_interface_name

__all__ = ["make_logger", "get_tcp_interface_name"]
