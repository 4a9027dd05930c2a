"""The driver depends on bench.py's JSON contract — lock it with a test."""
import json
import subprocess
import sys


def _free_port() -> int:
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=300,
    )
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
                "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config"):
        assert key in d, f"missing key {key}"
    assert d["metric"].startswith("training tokens/sec, Transformer-DMoE 256-expert")
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["value"] > 0 and d["steps"] == 2 and d["warmup"] == 1
    assert "global_batch" in d["config"] and "seq_len" in d["config"]


def test_bench_distributed_2proc_gloo():
    """The driver launches bench via torch.distributed.run for N>1 — lock
    that path end-to-end (EP all-to-all + averager + DHT control plane)."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()), "bench.py", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=600,
    )
    lines = [ln for ln in out.stdout.strip().splitlines() if ln.startswith("{")]
    assert lines, f"no JSON line; stderr tail: {out.stderr[-500:]}"
    d = json.loads(lines[-1])
    assert d["n_gpus"] == 2
    assert d["value"] > 0
    assert "ep2" in d["config"]["parallelism"]
