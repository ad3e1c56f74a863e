"""Shared 🤗 wrapper machinery for the task models.

Every task exposes the same three artifacts: a ``PretrainedConfig`` subclass
that carries the backend dataclass (serialized under ``model_config`` in
config.json), a ``PreTrainedModel`` holding the backend module as
``backend_model``, and a checkpoint-conversion entry point that turns a
training checkpoint into a ``save_pretrained`` directory. The reference
repeats that boilerplate per task; here it is factored once and the per-task
files declare only what differs (model_type, backend classes, pipeline glue).
"""
from __future__ import annotations

from dataclasses import asdict
from typing import Optional

from transformers import AutoTokenizer


class BackendConfigMixin:
    """``PretrainedConfig`` mixin that round-trips a backend dataclass.

    Subclasses set ``backend_config_class`` and may override
    ``default_backend_config`` (when the dataclass has required fields) or
    ``decode_backend_config`` (when nested dataclasses need rebuilding).
    """

    backend_config_class = None

    def __init__(self, backend_config=None, **kwargs):
        if backend_config is None:
            backend_config = self.default_backend_config()
        self.model_config = asdict(backend_config)
        super().__init__(**kwargs)

    @classmethod
    def default_backend_config(cls):
        return cls.backend_config_class()

    @classmethod
    def decode_backend_config(cls, model_config: dict):
        return cls.backend_config_class.create(**model_config)

    @property
    def backend_config(self):
        return self.decode_backend_config(self.model_config)


def wrap_lit_checkpoint(lit_class, hf_model_class, ckpt_path, *, is_decoder: bool):
    """Load a training checkpoint and wrap its backend model for 🤗 use.

    The state dict moves verbatim — the Lit wrapper, the backend module and
    the 🤗 wrapper all share the same ``encoder.*``/``decoder.*`` key layout.
    """
    backend = lit_class.load_from_checkpoint(ckpt_path).model
    config = hf_model_class.config_class(backend.config)
    config.is_decoder = is_decoder
    wrapper = hf_model_class(config)
    wrapper.backend_model.load_state_dict(backend.state_dict())
    return wrapper


def save_with_tokenizer(model, tokenizer_name: str, save_dir, *,
                        padding_side: Optional[str] = None,
                        config_overrides: Optional[dict] = None, **kwargs):
    """save_pretrained the model together with its tokenizer.

    Records the tokenizer class in the model config (so ``pipeline()`` picks
    it up) and applies any extra config attributes (id2label etc.).
    """
    tok_kwargs = {"verbose": False}
    if padding_side is not None:
        tok_kwargs["padding_side"] = padding_side
    tokenizer = AutoTokenizer.from_pretrained(tokenizer_name, **tok_kwargs)
    tokenizer.save_pretrained(save_dir, **kwargs)

    model.config.tokenizer_class = type(tokenizer).__name__
    for key, value in (config_overrides or {}).items():
        setattr(model.config, key, value)
    model.save_pretrained(save_dir, **kwargs)
