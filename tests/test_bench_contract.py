"""Driver-contract test for bench.py.

The round-end driver launches `python bench.py --gpus N --steps K
--warmup W` and parses ONE JSON line from rank 0; these keys are a hard
interface.  Runs the CPU config (no GPU needed) with a tiny step count."""

import json
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--config", "cpu",
         "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=300, cwd=str(REPO))
    assert out.returncode == 0, out.stderr[-800:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, key
    assert d["n_gpus"] == 1
    assert d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert isinstance(d["value"], (int, float)) and d["value"] > 0
    assert isinstance(d["ms_per_step"], (int, float)) and d["ms_per_step"] > 0
    cfg = d["config"]
    for key in ("model", "global_batch", "seq_len", "parallelism",
                "sessions", "payload_bytes"):
        assert key in cfg, key
    assert cfg["parallelism"] == "dp1"
