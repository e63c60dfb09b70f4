"""Training harness: native loop, checkpointing, evaluation."""

from .helpers import (  # noqa: F401
    count_parameters,
    fix_random_seed,
    get_exponential_decay_gamma,
    load_all_results,
    make_Xy_input,
    set_seed,
)
from .trainer import (  # noqa: F401
    CVSplit,
    NPFTrainer,
    eval_loglike,
    predefined_split,
    train_models,
)
