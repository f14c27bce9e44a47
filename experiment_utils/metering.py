from stochastic_gradient_push_amd.utils.metering import Meter

__all__ = ["Meter"]
