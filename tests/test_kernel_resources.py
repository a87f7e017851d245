"""Kernel-resource regression checks (CPU: hipcc cross-compiles gfx950).

Guards the performance envelope: register spills or scratch in the hot kernels
are silent multi-x slowdowns (SURVEY.md §5 sanitizer/CI obligation — the
rocprofv3 counter evidence lives in profiles/SUMMARY.md; this test pins the
compile-time side).
"""

import re
import shutil
import subprocess
from pathlib import Path

import pytest

HIP_DIR = Path(__file__).resolve().parent.parent / "mpgcn_amd" / "ops" / "hip"

hipcc = shutil.which("hipcc")
pytestmark = pytest.mark.skipif(hipcc is None, reason="hipcc not available")


def _resource_report(src: str) -> list[dict]:
    out = subprocess.run(
        [hipcc, "--offload-arch=gfx950", "-O3", "-std=c++17",
         "-Rpass-analysis=kernel-resource-usage", "-c", str(HIP_DIR / src),
         "-o", "/dev/null"],
        capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    kernels = []
    cur = None
    for line in out.stderr.splitlines():
        m = re.search(r"Function Name: (\S+)", line)
        if m:
            cur = {"name": m.group(1)}
            kernels.append(cur)
        m = re.search(r"(\w[\w ]*\w) ?(?:\[bytes/lane\])?: (\d+)", line)
        if m and cur is not None:
            cur[m.group(1).strip()] = int(m.group(2))
    return kernels


@pytest.mark.parametrize("src,max_scratch", [
    ("axis_gemm.hip", 64),
    ("row_gemm.hip", 0),
    ("red_gemm.hip", 64),   # 32 B/lane from the guarded scalar-staging fallback
    ("lstm.hip", 0),
    ("elemwise.hip", 0),
])
def test_no_spills_in_hot_kernels(src, max_scratch):
    for k in _resource_report(src):
        assert k.get("VGPRs Spill", 0) == 0, (src, k["name"])
        assert k.get("ScratchSize [bytes/lane]", k.get("ScratchSize", 0)) <= max_scratch, (
            src, k["name"])


def test_lstm_fused_t7_no_spills():
    """The flagship sequence length (T=7) must stay spill-free; longer-T
    variants may spill slightly (they take the slab path in practice)."""
    for k in _resource_report("lstm_fused.hip"):
        if "ILi7EE" in k["name"]:
            assert k.get("VGPRs Spill", 0) == 0, k["name"]
            assert k.get("ScratchSize", 0) == 0, k["name"]
