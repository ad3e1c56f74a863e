"""Convert official deepmind HF Perceiver models and trained checkpoints into this
framework's persistent 🤗 directories (parity: reference examples/convert.py).

Requires network access / local copies of the source models.
"""
import argparse


def checkpoint_url(run: str, name: str) -> str:
    return f"https://martin-krasser.com/perceiver/logs-0.8.0/{run}/checkpoints/{name}"


def main():
    p = argparse.ArgumentParser()
    p.add_argument("target", choices=[
        "mlm-model", "img-clf-model", "flow-model",
        "mlm-ckpt", "txt-clf-ckpt", "img-clf-ckpt", "clm-ckpt", "sam-ckpt",
    ])
    p.add_argument("--save_dir", required=True)
    p.add_argument("--source", default=None, help="source repo id or checkpoint path/url")
    args = p.parse_args()

    if args.target == "mlm-model":
        from perceiver_amd.models.text.mlm_hf import convert_model

        convert_model(args.save_dir, source_repo_id=args.source or "deepmind/language-perceiver")
    elif args.target == "img-clf-model":
        from perceiver_amd.models.vision.image_classifier_hf import convert_model

        convert_model(args.save_dir, source_repo_id=args.source or "deepmind/vision-perceiver-fourier")
    elif args.target == "flow-model":
        from perceiver_amd.models.vision.optical_flow_hf import convert_model

        convert_model(args.save_dir, source_repo_id=args.source or "deepmind/optical-flow-perceiver")
    elif args.target == "mlm-ckpt":
        from perceiver_amd.models.text.mlm_hf import convert_checkpoint

        convert_checkpoint(args.save_dir, args.source, tokenizer_name="deepmind/language-perceiver")
    elif args.target == "txt-clf-ckpt":
        from perceiver_amd.models.text.classifier_hf import convert_imdb_classifier_checkpoint

        convert_imdb_classifier_checkpoint(args.save_dir, args.source,
                                           tokenizer_name="deepmind/language-perceiver")
    elif args.target == "img-clf-ckpt":
        from perceiver_amd.models.vision.image_classifier_hf import convert_mnist_classifier_checkpoint

        convert_mnist_classifier_checkpoint(args.save_dir, args.source)
    elif args.target == "clm-ckpt":
        from perceiver_amd.models.text.clm_hf import convert_checkpoint

        convert_checkpoint(args.save_dir, args.source, tokenizer_name="deepmind/language-perceiver")
    elif args.target == "sam-ckpt":
        from perceiver_amd.models.audio.symbolic_hf import convert_checkpoint

        convert_checkpoint(args.save_dir, args.source)


if __name__ == "__main__":
    main()
