from cpd_amd.parallel import (DistModule, broadcast_params, dist_init,
                              kahan_sum_gradients, normal_sum_gradients,
                              sum_gradients)

__all__ = ["DistModule", "broadcast_params", "dist_init", "sum_gradients",
           "normal_sum_gradients", "kahan_sum_gradients"]
