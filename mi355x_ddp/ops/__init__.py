from . import _backend  # noqa: F401
from ._backend import extension_available, native_enabled  # noqa: F401
from .batchnorm import bn_relu, bn_add_relu  # noqa: F401
from .xent import softmax_cross_entropy  # noqa: F401
from .sgd import FusedSGD  # noqa: F401
from .conv import MI355Conv2d, conv2d  # noqa: F401
from .pool import GlobalAvgPool2d  # noqa: F401
