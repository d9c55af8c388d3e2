from .efficientnet import *  # noqa: F401,F403
from .factory import create_model  # noqa: F401
from .helpers import load_checkpoint, resume_checkpoint  # noqa: F401
from .registry import (is_model, list_models, list_modules,  # noqa: F401
                       model_entrypoint, register_model)
