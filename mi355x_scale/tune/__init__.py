"""W2: hyperopt-compatible distributed hyperparameter search.

Surface parity:
    from mi355x_scale.tune import fmin, tpe, hp, scope, GPUTrials,
                                  SparkTrials, Trials, STATUS_OK
"""

from .space import (hp, scope, bind_params, flatten_space,  # noqa: F401
                    space_eval)
from .tpe import tpe, rand, TPE  # noqa: F401
from .fmin import (fmin, Trials, GPUTrials, SparkTrials,  # noqa: F401
                   STATUS_OK, STATUS_FAIL)
