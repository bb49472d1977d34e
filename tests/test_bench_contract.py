"""The driver depends on bench.py's JSON line contract — lock it down."""
import json
import subprocess
import sys


def test_bench_json_contract():
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--ops", "torch_ref"],
        capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    rec = json.loads(r.stdout.strip().splitlines()[-1])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in rec, key
    assert rec["metric"] == "env_steps_per_sec"
    assert rec["higher_is_better"] is True
    assert rec["scaling"] == "strong"
    assert rec["data"] == "synthetic"
    assert rec["value"] > 0
    cfgk = rec["config"]
    for key in ("model", "global_batch", "seq_len", "parallelism"):
        assert key in cfgk, key
