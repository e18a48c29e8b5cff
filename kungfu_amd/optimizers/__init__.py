from kungfu_amd.optimizers.ada_sgd import AdaptiveSGDOptimizer  # noqa
from kungfu_amd.optimizers.async_sgd import PairAveragingOptimizer  # noqa
from kungfu_amd.optimizers.core import KungFuOptimizer  # noqa
from kungfu_amd.optimizers.grad_variance import (  # noqa
    MonitorGradientVarianceOptimizer)
from kungfu_amd.optimizers.noise_scale import (  # noqa
    GradNoiseScaleProbe, MonitorGradientNoiseScaleOptimizer)
from kungfu_amd.optimizers.sma_sgd import (  # noqa
    SynchronousAveragingOptimizer)
from kungfu_amd.optimizers.sync_sgd import SynchronousSGDOptimizer  # noqa
