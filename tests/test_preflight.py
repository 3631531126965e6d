"""Preflight checker tests: simulated healthy/broken node trees via
the override envs."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SCRIPT = os.path.join(REPO, "scripts", "preflight.py")


def run(env_extra):
    env = dict(os.environ)
    env.update(env_extra)
    r = subprocess.run([sys.executable, SCRIPT, "--json"],
                       env=env, capture_output=True, text=True,
                       timeout=60)
    return r.returncode, {c["check"]: c for c in json.loads(r.stdout)}


def make_healthy_node(tmp_path, xnack=True):
    topo = tmp_path / "nodes"
    cpu = topo / "0"; gpu = topo / "1"
    cpu.mkdir(parents=True); gpu.mkdir()
    (cpu / "properties").write_text("simd_count 0\ncapability 0\n")
    cap = 0x4 if xnack else 0
    (gpu / "properties").write_text(
        f"simd_count 1024\ncapability {cap}\n")
    kfd = tmp_path / "kfd"; kfd.write_text("")
    dri = tmp_path / "dri"; dri.mkdir()
    (dri / "renderD128").write_text("")
    return {"VGPU_KFD_TOPO_OVERRIDE": str(topo),
            "VGPU_DEV_KFD_OVERRIDE": str(kfd),
            "VGPU_DEV_DRI_OVERRIDE": str(dri),
            "HSA_ENABLE_IPC_MODE_LEGACY": "0"}


def test_healthy_node_passes(tmp_path):
    rc, checks = run(make_healthy_node(tmp_path))
    assert rc == 0, checks
    assert checks["kfd_gpu_nodes"]["status"] == "ok"
    assert checks["dev_kfd"]["status"] == "ok"
    assert checks["xnack"]["status"] == "ok"


def test_missing_gpu_fails(tmp_path):
    env = make_healthy_node(tmp_path)
    env["VGPU_KFD_TOPO_OVERRIDE"] = str(tmp_path / "nope")
    rc, checks = run(env)
    assert rc == 1
    assert checks["kfd_gpu_nodes"]["status"] == "fail"


def test_no_xnack_warns_not_fails(tmp_path):
    rc, checks = run(make_healthy_node(tmp_path, xnack=False))
    assert rc == 0
    assert checks["xnack"]["status"] == "warn"
