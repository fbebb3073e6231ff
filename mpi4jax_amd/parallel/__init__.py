from .comm import (  # noqa: F401
    Communicator,
    init,
    finalize,
    flush,
    get_default_comm,
    get_world,
    COMM_WORLD,
)
from .grid import CartesianGrid  # noqa: F401
from .ddp import average_gradients  # noqa: F401
