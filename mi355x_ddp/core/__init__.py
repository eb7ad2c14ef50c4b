from . import dist  # noqa: F401
from . import metrics  # noqa: F401
