"""Model factory.

Capability parity with reference dfd/timm/models/factory.py: `create_model`
(:8), `create_deepfake_model` (:67), `create_deepfake_model_v3` (:127),
`create_deepfake_model_v4` (:190-252, asserts the production model name).
"""

from .helpers import load_checkpoint
from .registry import is_model, is_model_in_modules, model_entrypoint

__all__ = [
    "create_model",
    "create_deepfake_model",
    "create_deepfake_model_v3",
    "create_deepfake_model_v4",
]


def _sanitize_kwargs(model_name, kwargs):
    """BN/drop kwargs are only supported by efficientnet/mobilenetv3 models;
    None-valued optional args are removed so other models don't break."""
    if not is_model_in_modules(model_name, ["efficientnet", "mobilenetv3"]):
        kwargs.pop("bn_tf", None)
        kwargs.pop("bn_momentum", None)
        kwargs.pop("bn_eps", None)

    if kwargs.get("drop_block_rate", None) is None:
        kwargs.pop("drop_block_rate", None)

    # backwards compat: drop_connect -> drop_path
    drop_connect_rate = kwargs.pop("drop_connect_rate", None)
    if drop_connect_rate is not None and kwargs.get("drop_path_rate", None) is None:
        kwargs["drop_path_rate"] = drop_connect_rate

    if kwargs.get("drop_path_rate", None) is None:
        kwargs.pop("drop_path_rate", None)
    return kwargs


def _create(model_name, margs, kwargs):
    if is_model(model_name):
        create_fn = model_entrypoint(model_name)
        return create_fn(**margs, **kwargs)
    raise RuntimeError("Unknown model (%s)" % model_name)


def create_model(model_name, pretrained=False, num_classes=1000, in_chans=3,
                 checkpoint_path="", ignore_keys=None, **kwargs):
    """Create any registered model by name."""
    margs = dict(pretrained=pretrained, num_classes=num_classes, in_chans=in_chans)
    kwargs = _sanitize_kwargs(model_name, kwargs)
    model = _create(model_name, margs, kwargs)
    if checkpoint_path:
        load_checkpoint(model, checkpoint_path, strict=False, ignore_keys=ignore_keys)
    return model


def create_deepfake_model(model_name, pretrained=False, num_classes=2, in_chans=3,
                          checkpoint_path="", ignore_keys=None, strict=False, **kwargs):
    margs = dict(pretrained=pretrained, num_classes=num_classes, in_chans=in_chans)
    kwargs = _sanitize_kwargs(model_name, kwargs)
    model = _create(model_name, margs, kwargs)
    if checkpoint_path:
        load_checkpoint(model, checkpoint_path, strict=strict, ignore_keys=ignore_keys)
    return model


def create_deepfake_model_v3(model_name, pretrained=False, num_classes=2, in_chans=3,
                             checkpoint_path="", ignore_keys=None, strict=False, **kwargs):
    assert model_name == "efficientnet_deepfake_v3", \
        "{} != efficientnet_deepfake_v3".format(model_name)
    return create_deepfake_model(
        model_name, pretrained=pretrained, num_classes=num_classes, in_chans=in_chans,
        checkpoint_path=checkpoint_path, ignore_keys=ignore_keys, strict=strict, **kwargs)


def create_deepfake_model_v4(model_name, pretrained=False, num_classes=2, in_chans=3,
                             checkpoint_path="", ignore_keys=None, strict=False, **kwargs):
    """Production-model factory: only `efficientnet_deepfake_v4` is accepted
    (reference factory.py:190-252)."""
    assert model_name == "efficientnet_deepfake_v4", \
        "{} != efficientnet_deepfake_v4".format(model_name)
    return create_deepfake_model(
        model_name, pretrained=pretrained, num_classes=num_classes, in_chans=in_chans,
        checkpoint_path=checkpoint_path, ignore_keys=ignore_keys, strict=strict, **kwargs)
