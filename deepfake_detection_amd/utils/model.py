"""Model unwrap helpers (reference dfd/timm/utils.py:25-33)."""


def unwrap_model(model):
    if hasattr(model, "module"):
        return model.module
    if hasattr(model, "ema"):
        return model.ema
    return model


def get_state_dict(model, unwrap_fn=unwrap_model):
    return unwrap_fn(model).state_dict()
