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
from .tp import (  # noqa: F401
    ColumnParallelLinear,
    RowParallelLinear,
    copy_to_parallel,
)
from .sp import seq_to_head_shard, head_to_seq_shard  # noqa: F401
from .pp import (  # noqa: F401
    send_activation,
    recv_activation,
    backward_send,
)
