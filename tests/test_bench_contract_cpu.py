"""Locks the driver's bench.py contract: one JSON line on stdout from
rank 0 with the exact keys/semantics BASELINE.json's runner expects."""
import json
import subprocess
import sys


def _run(args):
    r = subprocess.run([sys.executable, "bench.py"] + args,
                       capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    return json.loads(lines[0])


def test_bench_json_contract():
    j = _run(["--model", "tiny", "--steps", "2", "--warmup", "1"])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in j, key
    assert j["unit"] == "tokens/s" and j["higher_is_better"] is True
    assert j["scaling"] == "weak" and j["data"] == "synthetic"
    assert j["dtype"] == "bf16" and j["vs_baseline"] is None
    assert j["n_gpus"] == 1 and j["steps"] == 2 and j["warmup"] == 1
    assert j["value"] > 0 and j["ms_per_step"] > 0
    # whole-job aggregate: value [tok/s] * elapsed == batch * steps
    tokens = j["value"] * j["ms_per_step"] * j["steps"] / 1e3
    expect = j["config"]["global_batch"] * j["steps"]
    assert abs(tokens - expect) / expect < 0.02, (tokens, expect)
    cfg = j["config"]
    assert cfg["global_batch"] == 512 and cfg["parallelism"] == "tp1"


def test_bench_json_moe_metric_name():
    j = _run(["--model", "tiny-moe", "--steps", "2", "--warmup", "1"])
    assert j["metric"] == "tiny_moe_tp_decode_tokens_per_s"


def test_doctor_runs():
    """`python -m triton_dist_amd.doctor` exits 0 on this host."""
    r = subprocess.run([sys.executable, "-m", "triton_dist_amd.doctor"],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "extension: LOADED" in r.stdout, r.stdout
