"""Model registry: name -> entrypoint fn.

Capability parity with the reference registry (reference
dfd/timm/models/registry.py:14-94): `@register_model` decorator, name
listing with fnmatch filtering, module membership queries.
"""

import fnmatch
import re
import sys
from collections import defaultdict

__all__ = [
    "register_model",
    "list_models",
    "is_model",
    "model_entrypoint",
    "list_modules",
    "is_model_in_modules",
]

_module_to_models = defaultdict(set)  # module name -> set of model names
_model_to_module = {}  # model name -> module name
_model_entrypoints = {}  # model name -> entrypoint fn


def register_model(fn):
    """Decorator: register `fn` as the entrypoint for model `fn.__name__`."""
    mod = sys.modules[fn.__module__]
    module_name = fn.__module__.rsplit(".", 1)[-1]
    model_name = fn.__name__

    # add entrypoint fn to the module's __all__
    if hasattr(mod, "__all__"):
        if model_name not in mod.__all__:
            mod.__all__.append(model_name)
    else:
        mod.__all__ = [model_name]

    _model_entrypoints[model_name] = fn
    _model_to_module[model_name] = module_name
    _module_to_models[module_name].add(model_name)
    return fn


def _natural_key(s):
    return [int(t) if t.isdigit() else t for t in re.split(r"(\d+)", s.lower())]


def list_models(filter="", module=""):
    """Return sorted model names, optionally filtered by wildcard / module."""
    if module:
        models = list(_module_to_models[module])
    else:
        models = list(_model_entrypoints.keys())
    if filter:
        models = fnmatch.filter(models, filter)
    return sorted(models, key=_natural_key)


def is_model(model_name):
    return model_name in _model_entrypoints


def model_entrypoint(model_name):
    return _model_entrypoints[model_name]


def list_modules():
    return sorted(_module_to_models.keys())


def is_model_in_modules(model_name, module_names):
    assert isinstance(module_names, (tuple, list, set))
    return any(model_name in _module_to_models[n] for n in module_names)
