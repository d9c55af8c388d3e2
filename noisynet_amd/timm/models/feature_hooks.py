"""Forward-hook feature collector (reference timm/models/feature_hooks.py:31)."""

from collections import OrderedDict, defaultdict
from functools import partial


class FeatureHooks:
    def __init__(self, hooks, named_modules):
        modules = {k: v for k, v in named_modules}
        for h in hooks:
            hook_name = h['name']
            m = modules[hook_name]
            hook_fn = partial(self._collect_output_hook, hook_name)
            if h.get('type', 'forward') == 'forward_pre':
                m.register_forward_pre_hook(hook_fn)
            else:
                m.register_forward_hook(hook_fn)
        self._feature_outputs = defaultdict(OrderedDict)

    def _collect_output_hook(self, name, *args):
        x = args[-1]  # tensor we want is last argument, output for fwd hooks
        if isinstance(x, tuple):
            x = x[0]
        self._feature_outputs[x.device][name] = x

    def get_output(self, device):
        output = tuple(self._feature_outputs[device].values())
        self._feature_outputs[device] = OrderedDict()
        return output
