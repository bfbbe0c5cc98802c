from . import instructions  # noqa: F401
from .comm import GradReducer, Topology, init_topology  # noqa: F401
from .schedules import (  # noqa: F401
    SCHEDULES,
    GPipeSchedule,
    InferenceSchedule,
    NaiveParallelSchedule,
    PipeDreamFlushSchedule,
    Schedule,
)
from .worker import Worker  # noqa: F401
from .tp import (  # noqa: F401
    TPMLP,
    ColumnParallelLinear,
    RowParallelLinear,
    tp_mlp_layers,
)
