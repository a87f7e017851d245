"""bench.py driver contract: one JSON line with the mandated fields."""

import json
import subprocess
import sys


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "0",
         "--nodes", "16", "--batch", "2", "--device", "cpu"],
        capture_output=True, text=True, timeout=300, check=True,
    )
    line = out.stdout.strip().splitlines()[-1]
    j = json.loads(line)
    for field in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                  "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                  "dtype", "data", "config"):
        assert field in j, field
    assert j["metric"] == "train_samples_per_sec"
    assert j["higher_is_better"] is True
    assert j["scaling"] == "weak"
    assert j["data"] == "synthetic"
    assert j["config"]["model"] == "MPGCN"
    assert j["config"]["global_batch"] == 2
    assert j["value"] > 0


def test_tool_scripts_parse():
    """The GPU-only probe/bench tools must at least stay syntactically valid
    (they are exercised on gpurun boxes, not in the CPU suite)."""
    import ast
    from pathlib import Path

    repo = Path(__file__).resolve().parent.parent
    for rel in ("tools/det_probe.py", "tools/profile_step.py",
                "tools/fp8_probe.py", "bench_kernels.py", "bench_one.py",
                "mpgcn_amd/serve.py", "__graft_entry__.py"):
        src = (repo / rel).read_text()
        ast.parse(src)
