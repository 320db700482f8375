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


def test_bench_torchrun_world2_cpu(tmp_path):
    """bench.py under the DRIVER'S exact launch form (torch.distributed.run
    --nnodes=1 --nproc-per-node N ... bench.py --gpus N) at N=2 over
    gloo/CPU with a tiny model: the multi-rank path (init, sharding,
    barriers, MAX-over-ranks timing, rank-0 single JSON line) must work
    before the round-end 8-GPU SCALE run ever touches it."""
    import json
    import subprocess
    import sys
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29731", "bench.py", "--gpus", "2", "--steps",
         "2", "--warmup", "1", "--model", "llama2_125m", "--batch-size",
         "1", "--seq-len", "128", "--sharding", "fsdp"],
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        env=env, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(line) == 1, out.stdout   # exactly one JSON line (rank 0)
    d = json.loads(line[0])
    assert d["n_gpus"] == 2 and d["steps"] == 2
    assert d["config"]["parallelism"] == "fsdp2"
