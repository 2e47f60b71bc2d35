from .flat import FlatParams  # noqa: F401
from .ddp import FlatDDP  # noqa: F401
from .optim import FlatAdamW  # noqa: F401
