"""CLI: python -m perceiver_amd.serve --model <dir-or-ckpt> [--port 8000]."""
import argparse

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", required=True,
                    help="save_pretrained directory or Lightning .ckpt of a causal LM")
    ap.add_argument("--tokenizer", default=None,
                    help="tokenizer name/dir (default: PerceiverTokenizer byte-level)")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--bf16", action="store_true", default=None,
                    help="cast to bf16 (default on GPU)")
    args = ap.parse_args()

    from perceiver_amd.models.text.clm_hf import PerceiverCausalLanguageModel
    from perceiver_amd.serve import create_app

    if args.model.endswith(".ckpt"):
        model = PerceiverCausalLanguageModel.from_checkpoint(args.model)
    else:
        model = PerceiverCausalLanguageModel.from_pretrained(args.model)
    model.eval()
    if torch.cuda.is_available():
        dtype = torch.bfloat16 if args.bf16 in (None, True) else torch.float32
        model = model.to("cuda", dtype)

    if args.tokenizer:
        from transformers import AutoTokenizer

        tokenizer = AutoTokenizer.from_pretrained(args.tokenizer)
    else:
        from transformers import PerceiverTokenizer

        tokenizer = PerceiverTokenizer()

    import uvicorn

    uvicorn.run(create_app(model, tokenizer), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
