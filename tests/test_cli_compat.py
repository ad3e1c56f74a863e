"""Reference-invocation compatibility of the CLI config layer: the exact flag
style of the reference's examples (LightningCLI/jsonargparse) must parse and
map onto the native TrainConfig — class_path/init_args YAML, --trainer.*
Lightning flags (precision 16 etc.), --data=<Class> selection, --optimizer=
AdamW and --lr_scheduler.* sections (reference examples/training/*/train.sh)."""
import yaml

from perceiver_amd.scripts.cli import parse_cli_config, translate_lightning_trainer


def test_reference_mlm_train_sh_flags_map():
    # the flag set of reference examples/training/mlm/train.sh, verbatim
    argv = [
        "fit",
        "--model.params=krasserm/perceiver-io-mlm",
        "--model.activation_checkpointing=true",
        "--data=ImdbDataModule",
        "--data.tokenizer=krasserm/perceiver-io-mlm",
        "--data.add_special_tokens=true",
        "--data.static_masking=false",
        "--data.max_seq_len=2048",
        "--data.batch_size=32",
        "--optimizer=AdamW",
        "--optimizer.lr=1e-5",
        "--lr_scheduler.warmup_steps=1000",
        "--trainer.max_epochs=12",
        "--trainer.accelerator=gpu",
        "--trainer.precision=16",
        "--trainer.devices=2",
        "--trainer.log_every_n_steps=20",
        "--trainer.logger=TensorBoardLogger",
        "--trainer.logger.save_dir=logs",
        "--trainer.logger.name=mlm",
    ]
    sub, cfg = parse_cli_config(argv, defaults={})
    assert sub == "fit"
    assert cfg["data"]["_class_name"] == "ImdbDataModule"
    assert cfg["data"]["max_seq_len"] == 2048 and cfg["data"]["batch_size"] == 32
    assert cfg["model"]["activation_checkpointing"] is True
    assert cfg["optimizer"]["lr"] == 1e-5
    assert cfg["optimizer"]["optimizer"] == "adamw"
    assert cfg["optimizer"]["warmup_steps"] == 1000
    t = cfg["trainer"]
    assert t["precision"] == "fp16"       # Lightning "16" -> fp16 autocast
    assert t["max_epochs"] == 12
    assert t["log_every"] == 20
    assert "devices" not in t and "logger" not in t and "accelerator" not in t


def test_class_path_init_args_yaml(tmp_path):
    # the reference CLI's saved config.yaml format
    config = {
        "model": {
            "class_path": "perceiver.model.text.mlm.LitMaskedLanguageModel",
            "init_args": {"num_latents": 64, "num_latent_channels": 128},
        },
        "trainer": {
            "precision": "bf16-mixed",
            "gradient_clip_val": 0.5,
            "strategy": "ddp_find_unused_parameters_false",
            "max_steps": 100,
        },
    }
    f = tmp_path / "config.yaml"
    f.write_text(yaml.safe_dump(config))
    _, cfg = parse_cli_config(["fit", "--config", str(f)], defaults={})
    assert cfg["model"]["num_latents"] == 64
    assert cfg["trainer"]["precision"] == "bf16"
    assert cfg["trainer"]["grad_clip"] == 0.5
    assert cfg["trainer"]["max_steps"] == 100
    assert "strategy" not in cfg["trainer"]


def test_translate_precision_aliases():
    for raw, want in [(16, "fp16"), ("16-mixed", "fp16"), ("bf16", "bf16"),
                      (32, "fp32"), ("32-true", "fp32")]:
        assert translate_lightning_trainer({"precision": raw})["precision"] == want


def test_fp16_precision_trains_on_cpu():
    # fp16 requested but CPU: autocast/scaler disable themselves, training runs
    import torch

    from perceiver_amd.train.trainer import Trainer, TrainConfig

    torch.manual_seed(0)
    model = torch.nn.Linear(8, 4)
    t = Trainer(TrainConfig(max_steps=2, precision="fp16", log_every=100,
                            lr=1e-2, lr_schedule="none", out_dir="logs/_fp16"))
    t.fit_steps(model, [torch.randn(4, 8) for _ in range(2)],
                lambda m, b: m(b).square().mean())
    assert t.global_step == 2
