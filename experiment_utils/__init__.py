"""Experiment harness utilities (API parity with the reference's
experiment_utils package; implementations live in
stochastic_gradient_push_amd.utils)."""

from .cluster_manager import ClusterManager
from .helpers import get_tcp_interface_name, make_logger
from .metering import Meter

__all__ = ["ClusterManager", "get_tcp_interface_name", "make_logger", "Meter"]
