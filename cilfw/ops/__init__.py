from .functional import (conv2d, batchnorm_act, batchnorm_add_relu, add_relu,
                         downsample_a,
                         global_avg_pool, linear, cross_entropy, kd_loss, wa_loss,
                         accuracy, max_pool)
from ._backend import have_ext

__all__ = ["conv2d", "batchnorm_act", "batchnorm_add_relu", "add_relu",
           "downsample_a",
           "global_avg_pool", "linear", "cross_entropy", "kd_loss", "wa_loss", "accuracy",
           "max_pool", "have_ext"]
