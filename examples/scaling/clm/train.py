"""Perceiver-AR compute-optimal scaling study driver (parity: reference
examples/scaling/clm/train.py): sweeps channels/layers/steps on WikiText bytes.

    python examples/scaling/clm/train.py --num_channels 512 --num_layers 8 \
        --max_steps 25000 --max_seq_len 2048
"""
import argparse
import sys

sys.path.insert(0, ".")

from examples.scaling.clm.flops import ComputeEstimator  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--num_channels", type=int, default=512)
    p.add_argument("--num_layers", type=int, default=8)
    p.add_argument("--max_latents", type=int, default=512)
    p.add_argument("--max_seq_len", type=int, default=2048)
    p.add_argument("--max_steps", type=int, default=25000)
    p.add_argument("--batch_size", type=int, default=20)
    p.add_argument("--out_dir", default=None)
    p.add_argument("--dry_run", action="store_true", help="print the FLOPs estimate only")
    args = p.parse_args()

    est = ComputeEstimator(
        vocab_size=262, max_seq_len=args.max_seq_len, max_latents=args.max_latents,
        num_channels=args.num_channels, num_layers=args.num_layers,
    )
    total_seqs = args.max_steps * args.batch_size
    print(f"params (non-emb): {est.params()/1e6:.1f}M  "
          f"train FLOPs: {est.train_flops(total_seqs):.3e}")
    if args.dry_run:
        return

    from perceiver_amd.scripts.text.clm import CLI, DEFAULTS, WikiTextDataModule, build_model, link
    from perceiver_amd.train.lit import LitCausalLanguageModel

    out_dir = args.out_dir or (
        f"logs/scaling/c{args.num_channels}-l{args.num_layers}-s{args.max_steps}"
    )
    argv = [
        "fit",
        "--model.num_channels", str(args.num_channels),
        "--model.num_self_attention_layers", str(args.num_layers),
        "--model.max_latents", str(args.max_latents),
        "--data.max_seq_len", str(args.max_seq_len),
        "--data.batch_size", str(args.batch_size),
        "--trainer.max_steps", str(args.max_steps),
        "--trainer.out_dir", out_dir,
        "--optimizer.lr_schedule", "cosine",
    ]
    CLI(LitCausalLanguageModel, WikiTextDataModule, DEFAULTS, build_model, link, argv=argv)


if __name__ == "__main__":
    main()
