from bflc_amd.ops import functional
from bflc_amd.ops.functional import (accuracy, accuracy_t, adam_step_,
                                     axpy_, batchnorm2d, conv2d, conv2d_bn,
                                     global_avgpool, add_relu,
                                     hip_available, hip_ops, linear,
                                     maxpool2d, relu, sgd_step_,
                                     softmax_cross_entropy, weighted_fedavg)

__all__ = [
    "functional", "linear", "conv2d", "conv2d_bn", "batchnorm2d", "relu",
    "add_relu", "maxpool2d", "global_avgpool", "softmax_cross_entropy",
    "accuracy", "accuracy_t", "axpy_", "sgd_step_", "adam_step_",
    "weighted_fedavg", "hip_ops", "hip_available",
]
