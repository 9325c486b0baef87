from .optimizer import (  # noqa: F401
    Optimizer,
    SGD,
    Adam,
    LBSGD,
    create,
    get_updater,
    Updater,
)
