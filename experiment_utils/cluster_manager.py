from stochastic_gradient_push_amd.utils.cluster_manager import ClusterManager

__all__ = ["ClusterManager"]
