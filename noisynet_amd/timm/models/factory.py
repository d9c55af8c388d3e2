"""create_model dispatch (reference timm/models/factory.py:5-45)."""

from .helpers import load_checkpoint
from .registry import is_model, model_entrypoint


def create_model(model_name, pretrained=False, num_classes=1000, in_chans=3,
                 checkpoint_path='', **kwargs):
    margs = dict(pretrained=pretrained, num_classes=num_classes,
                 in_chans=in_chans)
    kwargs = {k: v for k, v in kwargs.items() if v is not None}
    if is_model(model_name):
        create_fn = model_entrypoint(model_name)
        model = create_fn(**margs, **kwargs)
    else:
        raise RuntimeError('Unknown model (%s)' % model_name)
    if checkpoint_path:
        load_checkpoint(model, checkpoint_path)
    return model
