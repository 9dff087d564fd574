from .callbacks import same_seeds, EarlyStopping   # noqa: F401
from .lookahead import Lookahead                    # noqa: F401
from .engine import iterate_loader                  # noqa: F401
from .train import train                            # noqa: F401
from .schedulers import WarmupLambdaLR, ReduceLROnPlateau  # noqa: F401
from .tboard import SummaryWriter                   # noqa: F401
