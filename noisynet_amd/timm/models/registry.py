"""Model registry (reference timm/models/registry.py:14-94)."""

import fnmatch
import re
import sys
from collections import defaultdict

_module_to_models = defaultdict(set)
_model_to_module = {}
_model_entrypoints = {}
_model_has_pretrained = set()


def register_model(fn):
    mod = sys.modules[fn.__module__]
    module_name_split = fn.__module__.split('.')
    module_name = module_name_split[-1] if len(module_name_split) else ''
    model_name = fn.__name__
    if hasattr(mod, '__all__'):
        mod.__all__.append(model_name)
    else:
        mod.__all__ = [model_name]
    _model_entrypoints[model_name] = fn
    _model_to_module[model_name] = module_name
    _module_to_models[module_name].add(model_name)
    has_pretrained = False
    if hasattr(mod, 'default_cfgs') and model_name in mod.default_cfgs:
        has_pretrained = 'url' in mod.default_cfgs[model_name] and \
            bool(mod.default_cfgs[model_name]['url'])
    if has_pretrained:
        _model_has_pretrained.add(model_name)
    return fn


def _natural_key(string_):
    return [int(s) if s.isdigit() else s for s in re.split(r'(\d+)', string_.lower())]


def list_models(filter='', module=''):
    models = _module_to_models[module] if module else _model_entrypoints.keys()
    if filter:
        models = fnmatch.filter(models, filter)
    return list(sorted(models, key=_natural_key))


def is_model(model_name):
    return model_name in _model_entrypoints


def model_entrypoint(model_name):
    return _model_entrypoints[model_name]


def list_modules():
    return list(sorted(_module_to_models.keys()))


def is_model_in_modules(model_name, module_names):
    return any(model_name in _module_to_models[n] for n in module_names)
