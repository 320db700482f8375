"""The driver's bench.py contract: `python bench.py --gpus N --steps K
--warmup W` must emit ONE JSON line with the agreed keys. This runs the
real script end-to-end on CPU with a tiny model so contract regressions
are caught before a GPU round."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--model", "llama2_test", "--gpus", "1", "--steps", "2",
         "--warmup", "1", "--batch-size", "1", "--seq-len", "64"],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in d, key
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    # aggregate value = per-GPU x N contract
    cfg = d["config"]
    assert cfg["global_batch"] == 1 and cfg["seq_len"] == 64
    assert abs(cfg["tok_per_sec_per_gpu"] * d["n_gpus"] - d["value"]) < 1.0
