"""Driver-contract test for bench.py: the exact launch the round-end driver uses
(torch.distributed.run, one rank per device) must produce ONE valid JSON line
from rank 0 with the contracted fields. Runs the tiny config on CPU over gloo."""
import json
import os
import subprocess
import sys

import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
BENCH = os.path.join(os.path.dirname(HERE), "bench.py")


def _run_bench(extra, nproc=1, timeout=300):
    if nproc == 1:
        cmd = [sys.executable, BENCH]
    else:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={nproc}", "--master-addr", "127.0.0.1",
               "--master-port", "29733", BENCH, "--gpus", str(nproc)]
    out = subprocess.run(cmd + extra, capture_output=True, text=True, timeout=timeout)
    assert out.returncode == 0, out.stdout + out.stderr
    json_lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, f"expected ONE JSON line, got: {out.stdout!r}"
    return json.loads(json_lines[0])


@pytest.mark.parametrize("model,metric", [
    ("mlm", "train_samples_per_s_mlm_seq2048"),
    ("clm-decode", "decode_tok_per_s_perceiver_ar_8192ctx"),
])
def test_bench_json_contract_single(model, metric):
    rec = _run_bench(["--tiny", "--device", "cpu", "--model", model,
                      "--steps", "2", "--warmup", "1"])
    assert rec["metric"] == metric
    for field in ("value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
                  "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config"):
        assert field in rec, field
    assert rec["value"] > 0 and rec["n_gpus"] == 1 and rec["steps"] == 2
    assert rec["data"] == "synthetic"


def test_bench_json_contract_two_rank_gloo():
    rec = _run_bench(["--tiny", "--device", "cpu", "--steps", "2", "--warmup", "1"],
                     nproc=2)
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["value"] > 0


def test_bench_json_contract_eight_rank_gloo():
    """The exact code path the driver times at N=8 (one rank per GPU over
    RCCL) executed end-to-end here over 8 gloo CPU ranks — reducer, metric
    aggregation and the JSON contract all exercised at the full rank count."""
    rec = _run_bench(["--tiny", "--device", "cpu", "--steps", "1", "--warmup", "0"],
                     nproc=8, timeout=600)
    assert rec["n_gpus"] == 8
    assert rec["config"]["parallelism"] == "dp8"
    assert rec["value"] > 0
