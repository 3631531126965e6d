"""Runs the C multi-process shared-memory stress tests under pytest.

These are the GPU-independent cross-process correctness proofs (token
conservation, ledger consistency, seqlock tearing) — the reference's
library/test/nogpu suite re-imagined for this ABI.
"""
import os
import subprocess

import pytest


@pytest.mark.parametrize("binary", [
    "test_config_seqlock",
    "test_vmem_region_concurrency",
    "test_sm_node_shared",
])
def test_nogpu_binary(built_core, binary):
    path = os.path.join(built_core, binary)
    r = subprocess.run([path], capture_output=True, text=True, timeout=180)
    assert r.returncode == 0, f"{binary} failed:\n{r.stdout}\n{r.stderr}"
    assert "PASS" in r.stdout
