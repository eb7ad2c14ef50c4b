from .flat_ddp import FlatDDP  # noqa: F401
from .wrap import wrap_data_parallel, wrap_torch_ddp  # noqa: F401
