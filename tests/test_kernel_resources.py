"""Compile-time kernel-resource guard (runs on CPU: hipcc cross-compiles).

Catches scratch-spill regressions — a dynamically-indexed accumulator
(CDNA guide rule 20) silently costs 5x; this suite failed the build the
one time it happened during development (conv_halo wrw, 176 B/lane).
"""

import os
import re
import shutil
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CSRC = os.path.join(REPO, "dynamic_load_balance_distributeddnn_amd",
                    "ops", "csrc")

HIPCC = shutil.which("hipcc")


@pytest.mark.skipif(HIPCC is None, reason="hipcc not on PATH")
@pytest.mark.parametrize("src", ["conv.hip", "conv_halo.hip", "conv_gn.hip",
                                 "conv_grouped.hip", "groupnorm.hip",
                                 "layernorm.hip", "attention.hip",
                                 "softmax.hip", "pool.hip", "maxpool.hip",
                                 "sgd.hip"])
def test_no_scratch_spills(src, tmp_path):
    out = subprocess.run(
        [HIPCC, "--offload-arch=gfx950", "-O3", "-std=c++17",
         "-Rpass-analysis=kernel-resource-usage", "-c",
         os.path.join(CSRC, src), "-o", str(tmp_path / "k.o")],
        capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    spills = re.findall(r"ScratchSize \[bytes/lane\]:\s*(\d+)", out.stderr)
    assert spills, "expected resource remarks"
    bad = [s for s in spills if int(s) != 0]
    assert not bad, f"{src}: scratch spill detected: {spills}"
