from .adamw import AdamW  # noqa: F401
from .lookahead import Lookahead  # noqa: F401
from .nadam import Nadam  # noqa: F401
from .novograd import NovoGrad, NvNovoGrad  # noqa: F401
from .optim_factory import add_weight_decay, create_optimizer  # noqa: F401
from .radam import RAdam  # noqa: F401
from .rmsprop_tf import RMSpropTF  # noqa: F401
