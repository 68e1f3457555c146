"""Model registry — replaces the reference's ``eval(config['model']['name'])``
construction (ESR:train_ours_cnt_seq.py:762) with an explicit registry."""

from __future__ import annotations

_MODELS: dict[str, type] = {}


def register_model(name: str):
    def deco(cls):
        _MODELS[name] = cls
        return cls
    return deco


def get_model_cls(name: str) -> type:
    try:
        return _MODELS[name]
    except KeyError:
        raise KeyError(f"unknown model '{name}'; registered: {sorted(_MODELS)}")


def build_model(name: str, **kwargs):
    return get_model_cls(name)(**kwargs)


def list_models():
    return sorted(_MODELS)
