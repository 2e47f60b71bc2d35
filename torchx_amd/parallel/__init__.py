from .flat import FlatParams  # noqa: F401
from .ddp import FlatDDP  # noqa: F401
from .optim import FlatAdamW  # noqa: F401
from .fp8 import Fp8Linear, convert_to_fp8  # noqa: F401
from .zero1 import FlatZeRO1  # noqa: F401
