from bflc_amd.ops import functional
from bflc_amd.ops.functional import (accuracy, adam_step_, axpy_, conv2d,
                                     hip_available, hip_ops, linear,
                                     maxpool2d, relu, sgd_step_,
                                     softmax_cross_entropy, weighted_fedavg)

__all__ = [
    "functional", "linear", "conv2d", "relu", "maxpool2d",
    "softmax_cross_entropy", "accuracy", "axpy_", "sgd_step_", "adam_step_",
    "weighted_fedavg", "hip_ops", "hip_available",
]
