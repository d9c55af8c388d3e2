"""Feature-map capture via module forward hooks.

Capability parity with the reference's hook collector used by
``EfficientNetFeatures`` (timm/models/feature_hooks.py:31): given a list of
hook specs ``[{'name': module_name, 'type': 'forward'|'forward_pre'}, ...]``
and the model's ``named_modules()``, it records each hooked module's output
tensor per device; ``get_output(device)`` drains the captures in hook
order.
"""


class FeatureHooks:
    def __init__(self, hooks, named_modules):
        by_name = dict(named_modules)
        self._order = [spec['name'] for spec in hooks]
        self._captures = {}  # device -> {name: tensor}
        for spec in hooks:
            module = by_name[spec['name']]
            if spec.get('type', 'forward') == 'forward_pre':
                module.register_forward_pre_hook(
                    self._make_pre_hook(spec['name']))
            else:
                module.register_forward_hook(
                    self._make_fwd_hook(spec['name']))

    def _store(self, name, tensor):
        if isinstance(tensor, tuple):
            tensor = tensor[0]
        self._captures.setdefault(tensor.device, {})[name] = tensor

    def _make_fwd_hook(self, name):
        def hook(module, inputs, output):
            self._store(name, output)
        return hook

    def _make_pre_hook(self, name):
        def hook(module, inputs):
            self._store(name, inputs[0] if isinstance(inputs, tuple)
                        else inputs)
        return hook

    def get_output(self, device):
        grabbed = self._captures.pop(device, {})
        return tuple(grabbed[n] for n in self._order if n in grabbed)
