"""bench.py driver-contract smoke: runs on CPU, emits the JSON line."""
import json
import os
import subprocess
import sys


def test_bench_cpu_smoke():
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "bench.py", "--model", "llama-tiny", "--steps", "1",
         "--warmup", "0", "--micro-batch", "1", "--grad-accum", "1",
         "--seq-len", "64"],
        cwd=repo, env=env, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in rec, key
    assert rec["n_gpus"] == 1
    assert rec["unit"] == "tokens/s"
    assert rec["data"] == "synthetic"
    assert rec["value"] > 0
    assert rec["config"]["seq_len"] == 64
