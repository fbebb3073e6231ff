from .reduce_ops import Op, SUM, PROD, MIN, MAX, AVG, BAND, BOR, BXOR  # noqa: F401
from .allgather import allgather  # noqa: F401
from .allreduce import allreduce  # noqa: F401
from .alltoall import alltoall  # noqa: F401
from .barrier import barrier  # noqa: F401
from .bcast import bcast  # noqa: F401
from .gather import gather  # noqa: F401
from .recv import recv  # noqa: F401
from .reduce import reduce  # noqa: F401
from .reduce_scatter import reduce_scatter  # noqa: F401
from .scan import scan  # noqa: F401
from .scatter import scatter  # noqa: F401
from .send import send  # noqa: F401
from .sendrecv import sendrecv  # noqa: F401
