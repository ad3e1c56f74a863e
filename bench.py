"""Flagship benchmark: Perceiver-IO MLM (seq=2048, 201M params) training samples/s.

Driver contract:
  python bench.py --gpus N --steps K --warmup W
  (for N>1 launched via torch.distributed.run, one rank per GPU over RCCL)

W untimed warmup steps, then exactly K timed steps bracketed by barrier +
torch.cuda.synchronize on both sides; MAX step time over ranks; rank 0 prints ONE
JSON line. Synthetic data (random byte tokens, 15% masked labels), random-init
weights, bf16 compute.

Additional benchmarks (not the headline): --model clm-decode measures Perceiver-AR
KV-cached decode tok/s; --model img / --model flow the big-KV encoders.
"""
from __future__ import annotations

import argparse
import json
import os
import time

import torch
import torch.distributed as dist
import torch.nn.functional as F


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--model", default="mlm", choices=["mlm", "clm", "clm-decode", "img", "flow"])
    p.add_argument("--batch", type=int, default=0, help="per-GPU batch size (0 = model default)")
    p.add_argument("--device", default=None, help="override device (e.g. cpu for local testing)")
    p.add_argument("--tiny", action="store_true", help="tiny config for CPU plumbing tests")
    p.add_argument("--eager-decode", action="store_true",
                   help="clm-decode: host-driven per-token steps instead of hipGraph replay")
    p.add_argument("--ddp-impl", default="native", choices=["native", "torch"],
                   help="gradient reducer: perceiver_amd.parallel bucketed RCCL reducer or torch DDP")
    p.add_argument("--seed", type=int, default=17)
    return p.parse_args()


def setup_dist(args):
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = torch.cuda.is_available() if args.device is None else args.device.startswith("cuda")
    if use_cuda:
        from perceiver_amd.utils.tunableop import arm_tunableop

        arm_tunableop(rank)
    if world_size > 1:
        backend = "nccl" if use_cuda else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(backend=backend, rank=rank, world_size=world_size)
    if use_cuda:
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")
    return rank, world_size, device


def barrier_sync(device):
    if dist.is_initialized():
        dist.barrier()
    if device.type == "cuda":
        torch.cuda.synchronize(device)


class MLMBench:
    unit = "samples/s"
    metric = "train_samples_per_s_mlm_seq2048"
    scaling = "weak"
    higher_is_better = True

    def __init__(self, args, device, rank):
        from perceiver_amd.models.flagship import mlm_flagship
        from perceiver_amd.models.text.mlm import MaskedLanguageModel

        if args.tiny:
            from perceiver_amd.models.text.common import TextEncoderConfig
            from perceiver_amd.models.text.mlm import MaskedLanguageModelConfig, TextDecoderConfig

            cfg = MaskedLanguageModelConfig(
                encoder=TextEncoderConfig(vocab_size=262, max_seq_len=128, num_input_channels=64,
                                          num_cross_attention_heads=4, num_self_attention_heads=4,
                                          num_self_attention_layers_per_block=2),
                decoder=TextDecoderConfig(vocab_size=262, max_seq_len=128, num_cross_attention_heads=4),
                num_latents=32, num_latent_channels=64,
            )
            self.seq, self.batch = 128, args.batch or 4
        else:
            cfg = mlm_flagship()
            self.seq, self.batch = 2048, args.batch or 32

        torch.manual_seed(args.seed + rank)
        self.model = MaskedLanguageModel(cfg).to(device)
        self.device = device
        self.cfg = cfg
        self.pure_bf16 = device.type == "cuda"
        if self.pure_bf16:
            # pure-bf16 weights + fp32-master AdamW: no autocast cast traffic
            from perceiver_amd.train.optim import MasterAdamW, convert_to_bf16_training

            self.model = convert_to_bf16_training(self.model)
            self.opt = MasterAdamW(self.model.parameters(), lr=2e-4, weight_decay=0.01)
        else:
            self.opt = torch.optim.AdamW(self.model.parameters(), lr=2e-4, weight_decay=0.01,
                                         betas=(0.9, 0.999), foreach=True)
        self.model.train()

        # synthetic batch: random bytes, 15% positions carry labels (masked-LM style)
        g = torch.Generator(device="cpu").manual_seed(args.seed + rank)
        self.x = torch.randint(6, 262, (self.batch, self.seq), generator=g).to(device)
        self.pad = torch.zeros(self.batch, self.seq, dtype=torch.bool, device=device)
        labels = torch.randint(6, 262, (self.batch, self.seq), generator=g)
        mask = torch.rand(self.batch, self.seq, generator=g) > 0.15
        labels[mask] = -100
        self.labels = labels.to(device)

    def wrap_ddp(self, impl):
        if not dist.is_initialized():
            return
        if impl == "torch":
            self.model = torch.nn.parallel.DistributedDataParallel(
                self.model, gradient_as_bucket_view=True, static_graph=True,
            )
        else:
            from perceiver_amd.parallel import BucketedGradReducer

            self.reducer = BucketedGradReducer(self.model)

    def step(self):
        logits = self.model(self.x, self.pad)
        loss = F.cross_entropy(logits.flatten(0, 1).float(), self.labels.flatten())
        self.opt.zero_grad(set_to_none=True)
        loss.backward()
        if getattr(self, "reducer", None) is not None:
            self.reducer.finalize()
        self.opt.step()
        return loss

    def items_per_step(self, world):
        return self.batch * world  # samples

    def config_json(self):
        return {"model": "perceiver-io-mlm-201M", "global_batch": None, "seq_len": self.seq,
                "num_latents": 512, "parallelism": None}


class CLMTrainBench:
    """Perceiver-AR causal-LM training (the reference's WikiText-103 bytes
    flagship: ctx 4096, 512 latents, 512 ch, 9 layers, CA-dropout 0.5,
    examples/training/clm/train.sh): samples/s over ranks."""

    unit = "samples/s"
    metric = "train_samples_per_s_clm_wikitext4096"
    scaling = "weak"
    higher_is_better = True

    def __init__(self, args, device, rank):
        from perceiver_amd.models.flagship import clm_wikitext
        from perceiver_amd.models.text.clm import CausalLanguageModel, CausalLanguageModelConfig

        if args.tiny:
            cfg = CausalLanguageModelConfig(vocab_size=262, max_seq_len=128, max_latents=32,
                                            num_channels=64, num_heads=4,
                                            num_self_attention_layers=2)
            self.batch = args.batch or 2
        else:
            cfg = clm_wikitext()
            self.batch = args.batch or 24

        torch.manual_seed(args.seed + rank)
        self.model = CausalLanguageModel(cfg).to(device)
        self.device = device
        self.cfg = cfg
        if device.type == "cuda":
            from perceiver_amd.train.optim import MasterAdamW, convert_to_bf16_training

            self.model = convert_to_bf16_training(self.model)
            self.opt = MasterAdamW(self.model.parameters(), lr=2e-4, weight_decay=0.01)
        else:
            self.opt = torch.optim.AdamW(self.model.parameters(), lr=2e-4, foreach=True)
        self.model.train()
        self.prefix_len = cfg.max_seq_len - cfg.max_latents
        g = torch.Generator(device="cpu").manual_seed(args.seed + rank)
        self.x = torch.randint(0, cfg.vocab_size, (self.batch, cfg.max_seq_len), generator=g).to(device)
        self.labels = self.x[:, self.prefix_len:].contiguous()

    wrap_ddp = MLMBench.wrap_ddp

    def step(self):
        out = self.model(self.x, prefix_len=self.prefix_len)
        loss = F.cross_entropy(out.logits.flatten(0, 1).float(), self.labels.flatten())
        self.opt.zero_grad(set_to_none=True)
        loss.backward()
        if getattr(self, "reducer", None) is not None:
            self.reducer.finalize()
        self.opt.step()
        return loss

    def items_per_step(self, world):
        return self.batch * world

    def config_json(self):
        return {"model": "perceiver-ar-30.7M-wikitext", "global_batch": None,
                "seq_len": self.cfg.max_seq_len, "max_latents": self.cfg.max_latents,
                "parallelism": None}


class CLMDecodeBench:
    """Perceiver-AR KV-cached decode: tokens/s aggregated over batch and ranks."""

    unit = "tok/s"
    metric = "decode_tok_per_s_perceiver_ar_8192ctx"
    scaling = "weak"
    higher_is_better = True

    def __init__(self, args, device, rank):
        from perceiver_amd.models.flagship import clm_flagship
        from perceiver_amd.models.text.clm import CausalLanguageModel

        if args.tiny:
            from perceiver_amd.models.text.clm import CausalLanguageModelConfig

            cfg = CausalLanguageModelConfig(vocab_size=262, max_seq_len=128, max_latents=32,
                                            num_channels=64, num_heads=4, num_self_attention_layers=2)
            self.batch = args.batch or 2
        else:
            cfg = clm_flagship()
            self.batch = args.batch or 32

        torch.manual_seed(args.seed + rank)
        self.model = CausalLanguageModel(cfg).to(device).eval()
        if device.type == "cuda":
            self.model = self.model.to(torch.bfloat16)
        self.device = device
        self.cfg = cfg
        self.prompt_len = cfg.max_seq_len - cfg.max_latents
        g = torch.Generator(device="cpu").manual_seed(args.seed + rank)
        self.prompt = torch.randint(0, cfg.vocab_size, (self.batch, self.prompt_len), generator=g).to(device)
        # decode_steps tokens generated per bench "step"
        self.decode_steps = 16 if not args.tiny else 4
        self.use_graph = device.type == "cuda" and not args.eager_decode

    def wrap_ddp(self, impl):
        pass  # inference benchmark: no gradient reduction

    @torch.no_grad()
    def step(self):
        # prefill once per step then decode_steps cached single-token steps,
        # using the preallocated in-place KV cache (no per-step concat);
        # on GPU the per-token step is a single hipGraph replay
        from perceiver_amd.core.cache import allocate_kv_cache

        if getattr(self, "_kv", None) is None:
            p = next(self.model.parameters())
            self._kv = allocate_kv_cache(self.model, self.batch, device=p.device, dtype=p.dtype)
            if self.use_graph:
                from perceiver_amd.core.graph_decode import GraphedDecoder

                self._gd = GraphedDecoder(self.model, self._kv)
        if self.use_graph:
            self._gd.prefill(self.prompt, prefix_len=self.prompt_len - 1)
            self._gd.decode(self.decode_steps - 1)
            return None
        for c in self._kv:
            c.reset()
        out = self.model(self.prompt, prefix_len=self.prompt_len - 1, kv_cache=self._kv)
        tok = out.logits[:, -1:].argmax(-1)
        for _ in range(self.decode_steps - 1):
            out = self.model(tok, prefix_len=0, kv_cache=self._kv)
            tok = out.logits[:, -1:].argmax(-1)
        return None

    def items_per_step(self, world):
        return self.batch * self.decode_steps * world  # generated tokens

    def config_json(self):
        return {"model": "perceiver-ar-8192ctx-1024lat", "global_batch": None,
                "seq_len": self.cfg.max_seq_len, "max_latents": self.cfg.max_latents,
                "decode_steps_per_bench_step": self.decode_steps, "parallelism": None}


class ImgBench(MLMBench):
    unit = "samples/s"
    metric = "train_samples_per_s_img_clf_224"
    scaling = "weak"
    higher_is_better = True

    def __init__(self, args, device, rank):
        from perceiver_amd.models.flagship import image_classifier_flagship
        from perceiver_amd.models.vision.image_classifier import ImageClassifier

        cfg = image_classifier_flagship()
        if args.tiny:
            from perceiver_amd.core import ClassificationDecoderConfig
            from perceiver_amd.models.vision.image_classifier import ImageClassifierConfig, ImageEncoderConfig

            cfg = ImageClassifierConfig(
                encoder=ImageEncoderConfig(image_shape=(16, 16, 3), num_frequency_bands=8,
                                           num_cross_attention_heads=1, num_self_attention_heads=2,
                                           num_self_attention_layers_per_block=2, num_self_attention_blocks=2),
                decoder=ClassificationDecoderConfig(num_classes=10, num_output_query_channels=32),
                num_latents=16, num_latent_channels=32,
            )
        self.batch = args.batch or (2 if args.tiny else 8)
        torch.manual_seed(args.seed + rank)
        self.model = ImageClassifier(cfg).to(device)
        self.device = device
        self.cfg = cfg
        self._setup_precision_and_opt()
        self.model.train()
        shape = cfg.encoder.image_shape
        g = torch.Generator(device="cpu").manual_seed(args.seed + rank)
        self.x = torch.randn(self.batch, *shape, generator=g).to(device)
        self.labels = torch.randint(0, cfg.decoder.num_classes, (self.batch,), generator=g).to(device)

    def _setup_precision_and_opt(self):
        # same scheme as MLMBench: pure-bf16 weights + fp32-master AdamW on GPU
        # (fused LN/dropout paths key on bf16 weights; no autocast cast traffic)
        self.pure_bf16 = self.device.type == "cuda"
        if self.pure_bf16:
            from perceiver_amd.train.optim import MasterAdamW, convert_to_bf16_training

            self.model = convert_to_bf16_training(self.model)
            self.opt = MasterAdamW(self.model.parameters(), lr=2e-4, weight_decay=0.01)
        else:
            self.opt = torch.optim.AdamW(self.model.parameters(), lr=2e-4, foreach=True)

    def step(self):
        x = self.x.to(torch.bfloat16) if self.pure_bf16 else self.x
        logits = self.model(x)
        loss = F.cross_entropy(logits.float(), self.labels)
        self.opt.zero_grad(set_to_none=True)
        loss.backward()
        if getattr(self, "reducer", None) is not None:
            self.reducer.finalize()
        self.opt.step()
        return loss

    def config_json(self):
        return {"model": "perceiver-io-img-clf-224-fourier64", "global_batch": None,
                "image_shape": list(self.cfg.encoder.image_shape), "parallelism": None}


class FlowBench(ImgBench):
    unit = "samples/s"
    metric = "train_samples_per_s_optical_flow_368x496"

    def __init__(self, args, device, rank):
        from perceiver_amd.models.flagship import optical_flow_flagship
        from perceiver_amd.models.vision.optical_flow import OpticalFlow

        cfg = optical_flow_flagship()
        if args.tiny:
            from perceiver_amd.models.vision.optical_flow import (
                OpticalFlowConfig, OpticalFlowDecoderConfig, OpticalFlowEncoderConfig,
            )

            cfg = OpticalFlowConfig(
                encoder=OpticalFlowEncoderConfig(image_shape=(16, 24), num_patch_input_channels=5,
                                                 num_patch_hidden_channels=8, num_frequency_bands=2,
                                                 num_cross_attention_heads=1, num_self_attention_heads=2,
                                                 num_self_attention_layers_per_block=2),
                decoder=OpticalFlowDecoderConfig(image_shape=(16, 24), num_cross_attention_heads=1),
                num_latents=16, num_latent_channels=32,
            )
        self.batch = args.batch or (1 if not args.tiny else 2)
        torch.manual_seed(args.seed + rank)
        self.model = OpticalFlow(cfg).to(device)
        self.device = device
        self.cfg = cfg
        self._setup_precision_and_opt()
        self.model.train()
        h, w = cfg.encoder.image_shape
        c = cfg.encoder.num_patch_input_channels
        g = torch.Generator(device="cpu").manual_seed(args.seed + rank)
        self.x = torch.randn(self.batch, 2, c, h, w, generator=g).to(device)
        self.target = torch.randn(self.batch, h, w, 2, generator=g).to(device)

    def step(self):
        x = self.x.to(torch.bfloat16) if self.pure_bf16 else self.x
        flow = self.model(x)
        loss = F.mse_loss(flow.float(), self.target)
        self.opt.zero_grad(set_to_none=True)
        loss.backward()
        if getattr(self, "reducer", None) is not None:
            self.reducer.finalize()
        self.opt.step()
        return loss

    def config_json(self):
        return {"model": "perceiver-io-optical-flow-368x496", "global_batch": None,
                "image_shape": list(self.cfg.encoder.image_shape), "parallelism": None}


BENCHES = {"mlm": MLMBench, "clm": CLMTrainBench, "clm-decode": CLMDecodeBench, "img": ImgBench, "flow": FlowBench}


def main():
    args = parse_args()
    rank, world, device = setup_dist(args)

    bench = BENCHES[args.model](args, device, rank)
    bench.wrap_ddp(args.ddp_impl)

    for _ in range(args.warmup):
        bench.step()

    barrier_sync(device)
    t0 = time.perf_counter()
    for _ in range(args.steps):
        bench.step()
    barrier_sync(device)
    elapsed = time.perf_counter() - t0

    # MAX step time over ranks == MIN throughput; reduce elapsed via all_reduce MAX
    if dist.is_initialized():
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1e3
    value = bench.items_per_step(world) * args.steps / elapsed

    if rank == 0:
        cfg = bench.config_json()
        cfg["global_batch"] = getattr(bench, "batch", 0) * world
        cfg["parallelism"] = f"dp{world}"
        print(json.dumps({
            "metric": bench.metric,
            "value": round(value, 3),
            "unit": bench.unit,
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": bench.higher_is_better,
            "scaling": bench.scaling,
            "vs_baseline": None,  # reference publishes no throughput numbers
            "dtype": "bf16" if device.type == "cuda" else "fp32",
            "data": "synthetic",
            "config": cfg,
        }))

    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
