from .schedule import NoamSchedule  # noqa: F401
from .optimizer import NoamAdam  # noqa: F401
from .metrics import Mean  # noqa: F401
from .summary import SummaryWriter  # noqa: F401
from .checkpoint import CheckpointManager  # noqa: F401
from .train_loop import Train, DistributedTrain  # noqa: F401
from .export import export_model, load_exported  # noqa: F401
