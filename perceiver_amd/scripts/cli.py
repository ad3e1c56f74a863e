"""Training CLI — the LightningCLI-equivalent of the reference's scripts/cli.py,
built on the native Trainer.

Configuration model:
  - per-task scripts supply flagship defaults (the reference's ``set_defaults``)
    and link functions (the reference's ``link_arguments``, e.g. data.vocab_size ->
    model.encoder.vocab_size);
  - flags use dotted paths: ``--model.num_latents 256 --data.batch_size 32
    --trainer.max_steps 1000 --optimizer.lr 1e-3``;
  - ``--config file.yaml`` merges a YAML tree under the same keys; later flags win;
  - the resolved config is saved to ``<out_dir>/config.yaml``.

Subcommands: fit (default), validate.
"""
from __future__ import annotations

import copy
import os
import sys
from dataclasses import fields, is_dataclass
from typing import Any, Callable, Dict, Optional

import yaml

from perceiver_amd.train.trainer import TrainConfig, Trainer


def _set_path(tree: Dict, path: str, value: Any) -> None:
    keys = path.split(".")
    node = tree
    for k in keys[:-1]:
        nxt = node.get(k)
        if not isinstance(nxt, dict):
            # LightningCLI style writes a class name at the section key
            # (--data=ImdbDataModule) before section fields; keep it aside
            node[k] = {"_class_name": nxt} if isinstance(nxt, str) else {}
            nxt = node[k]
        node = nxt
    last = keys[-1]
    if isinstance(node.get(last), dict) and not isinstance(value, dict):
        node[last]["_class_name"] = value
    else:
        node[last] = value


def _merge(base: Dict, override: Dict) -> Dict:
    out = copy.deepcopy(base)
    for k, v in override.items():
        if isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = _merge(out[k], v)
        else:
            out[k] = v
    return out


def _parse_value(raw: str) -> Any:
    try:
        value = yaml.safe_load(raw)
    except yaml.YAMLError:
        return raw
    if isinstance(value, str):
        # YAML 1.1 misses bare scientific notation ("1e-5" stays a string)
        try:
            return float(value) if any(c in value for c in ".eE") and value[0] in "0123456789+-." else value
        except ValueError:
            return value
    return value


def _strip_class_paths(node: Any) -> Any:
    """Flatten LightningCLI/jsonargparse ``{class_path, init_args}`` nodes.

    The reference's saved run configs (and its docs' examples) wrap
    instantiable sections in class_path/init_args; this CLI binds by section
    name, so the wrapper collapses to its init_args (recursively).
    """
    if isinstance(node, dict):
        if "class_path" in node and set(node) <= {"class_path", "init_args"}:
            return _strip_class_paths(node.get("init_args", {}))
        return {k: _strip_class_paths(v) for k, v in node.items()}
    if isinstance(node, list):
        return [_strip_class_paths(v) for v in node]
    return node


# Lightning Trainer flag -> native TrainConfig field (None = accepted, ignored)
_LIGHTNING_TRAINER_KEYS = {
    "gradient_clip_val": "grad_clip",
    "default_root_dir": "out_dir",
    "accumulate_grad_batches": "accumulate_grad_batches",
    "max_steps": "max_steps",
    "max_epochs": "max_epochs",
    "precision": "precision",
    "devices": None,      # process count comes from the torchrun launch
    "accelerator": None,
    "strategy": None,     # DDP is explicit here (BucketedGradReducer)
    "num_sanity_val_steps": None,
    "logger": None,
    "callbacks": None,
    "log_every_n_steps": "log_every",
    "val_check_interval": "val_every_steps",
}

_PRECISION_ALIASES = {
    "16": "fp16", "16-mixed": "fp16", "fp16": "fp16",
    "bf16": "bf16", "bf16-mixed": "bf16",
    "32": "fp32", "32-true": "fp32", "fp32": "fp32",
}


def translate_lightning_trainer(trainer_cfg: Dict) -> Dict:
    """Map the reference's Lightning trainer flags onto TrainConfig fields,
    so e.g. ``--trainer.precision 16 --trainer.gradient_clip_val 0.5``
    (examples/training/*/train.sh) run unchanged."""
    out: Dict = {}
    for key, value in trainer_cfg.items():
        target = _LIGHTNING_TRAINER_KEYS.get(key, key)
        if target is None:
            continue
        if target == "precision":
            value = _PRECISION_ALIASES.get(str(value), value)
        out[target] = value
    return out


def parse_cli_config(argv, defaults: Dict) -> (str, Dict):
    """Returns (subcommand, merged config tree)."""
    subcommand = "fit"
    if argv and not argv[0].startswith("-"):
        subcommand = argv[0]
        argv = argv[1:]

    tree: Dict = {}
    i = 0
    while i < len(argv):
        arg = argv[i]
        if not arg.startswith("--"):
            raise SystemExit(f"unexpected argument: {arg}")
        key = arg[2:]
        if "=" in key:
            key, raw = key.split("=", 1)
            i += 1
        else:
            if i + 1 >= len(argv):
                raise SystemExit(f"missing value for --{key}")
            raw = argv[i + 1]
            i += 2
        if key == "config":
            with open(raw) as f:
                loaded = _strip_class_paths(yaml.safe_load(f) or {})
            tree = _merge(tree, loaded)
        else:
            _set_path(tree, key, _parse_value(raw))

    tree = _merge(defaults, _strip_class_paths(tree))
    if "trainer" in tree:
        tree["trainer"] = translate_lightning_trainer(tree["trainer"])
    # reference-style --lr_scheduler.* section: fold into the optimizer section
    # (warmup + schedule selection live on TrainConfig here)
    lrs = tree.pop("lr_scheduler", None)
    if isinstance(lrs, dict):
        opt = tree.setdefault("optimizer", {})
        if "warmup_steps" in lrs:
            opt.setdefault("warmup_steps", lrs["warmup_steps"])
        sched_name = str(lrs.get("_class_name", ""))
        if "Cosine" in sched_name:
            opt.setdefault("lr_schedule", "cosine")
        elif "Constant" in sched_name:
            opt.setdefault("lr_schedule", "constant")
    # --optimizer=AdamW / Lamb class selection
    opt = tree.get("optimizer")
    if isinstance(opt, dict) and "_class_name" in opt:
        name = str(opt.pop("_class_name")).lower()
        opt.setdefault("optimizer", "lamb" if "lamb" in name else "adamw")
    return subcommand, tree


def build_dataclass(cls, cfg: Dict):
    """Instantiate a (possibly nested) dataclass from a config dict, ignoring
    unknown keys (mirrors jsonargparse's dataclass binding)."""
    if not is_dataclass(cls):
        return cfg
    kwargs = {}
    for f in fields(cls):
        if f.name in cfg:
            kwargs[f.name] = cfg[f.name]
    return cls(**kwargs)


def _resolve_datamodule(name: str):
    """LightningCLI-style class selection (--data=ImdbDataModule): look the
    name up across the data packages."""
    import importlib

    for pkg in ("perceiver_amd.data.text", "perceiver_amd.data.vision",
                "perceiver_amd.data.audio"):
        module = importlib.import_module(pkg)
        if hasattr(module, name):
            return getattr(module, name)
    raise SystemExit(f"unknown data module class: {name}")


class CLI:
    """Per-task training CLI driver.

    :param lit_cls: task wrapper class (perceiver_amd.train.lit.*)
    :param datamodule_cls: data module class
    :param defaults: {"model": {...}, "data": {...}, "trainer": {...},
                      "optimizer": {...}} flagship defaults
    :param build_model: callable(model_cfg: dict, datamodule) -> lit task instance
    :param link: callable(cfg, datamodule) mutating cfg["model"] from data properties
    """

    def __init__(self, lit_cls, datamodule_cls, defaults: Dict,
                 build_model: Callable, link: Optional[Callable] = None,
                 argv=None, run: bool = True):
        self.lit_cls = lit_cls
        self.datamodule_cls = datamodule_cls
        argv = sys.argv[1:] if argv is None else argv
        self.subcommand, self.config = parse_cli_config(argv, defaults)
        self.build_model = build_model
        self.link = link
        if run:
            self.run()

    def run(self):
        cfg = self.config
        data_cfg = dict(cfg.get("data", {}))
        class_name = data_cfg.pop("_class_name", None)
        dm_cls = self.datamodule_cls if class_name is None else _resolve_datamodule(class_name)
        if isinstance(cfg.get("model"), dict):
            cfg["model"].pop("_class_name", None)
        datamodule = dm_cls(**data_cfg)
        if self.link is not None:
            self.link(cfg, datamodule)

        trainer_cfg = dict(cfg.get("trainer", {}))
        opt_cfg = dict(cfg.get("optimizer", {}))
        for k in ("lr", "weight_decay", "warmup_steps", "lr_schedule", "min_lr_fraction", "optimizer"):
            if k in opt_cfg:
                trainer_cfg[k] = opt_cfg[k]
        ckpt_path = trainer_cfg.pop("ckpt_path", None)
        tc = TrainConfig(**trainer_cfg)
        trainer = Trainer(tc)

        task = self.build_model(cfg.get("model", {}), datamodule)

        from perceiver_amd.parallel import is_main_process

        if is_main_process():
            os.makedirs(tc.out_dir, exist_ok=True)
            with open(os.path.join(tc.out_dir, "config.yaml"), "w") as f:
                yaml.safe_dump(_sanitize(cfg), f)

        if self.subcommand == "fit":
            trainer.fit(task, datamodule=datamodule, ckpt_path=ckpt_path)
        elif self.subcommand == "validate":
            datamodule.prepare_data()
            datamodule.setup("validate")
            if ckpt_path:
                trainer._restore(task, None, None, ckpt_path)
            task = task.to(trainer.device)
            metrics = trainer._validate(task, datamodule.val_dataloader())
            print(metrics)
        else:
            raise SystemExit(f"unknown subcommand {self.subcommand}")
        return trainer


def _sanitize(obj):
    from dataclasses import asdict

    if is_dataclass(obj) and not isinstance(obj, type):
        return asdict(obj)
    if isinstance(obj, dict):
        return {k: _sanitize(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return [_sanitize(v) for v in obj]
    if isinstance(obj, (str, int, float, bool)) or obj is None:
        return obj
    return str(obj)
