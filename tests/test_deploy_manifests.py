"""Deploy manifest lint: YAML parses, every container command maps to
a real cmd module, and every flag the manifests pass actually exists
in that binary's argparse surface (catches manifest/CLI drift)."""
import glob
import os
import re
import subprocess
import sys

import pytest
import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
MANIFESTS = sorted(glob.glob(os.path.join(REPO, "deploy", "**", "*.yaml"),
                             recursive=True))


def iter_containers():
    for path in MANIFESTS:
        if "/chart/" in path:
            continue  # helm templates are not plain yaml
        for doc in yaml.safe_load_all(open(path)):
            if not isinstance(doc, dict):
                continue
            spec = doc.get("spec", {}) or {}
            pod = (spec.get("template", {}) or {}).get("spec") or \
                (spec if doc.get("kind") == "Pod" else None)
            if not pod:
                continue
            for c in pod.get("containers") or []:
                yield path, c


def test_all_manifests_parse():
    assert MANIFESTS
    for path in MANIFESTS:
        if "/chart/" in path:
            continue
        list(yaml.safe_load_all(open(path)))


@pytest.mark.parametrize("path,container", [
    (p, c) for p, c in iter_containers()
    if (c.get("command") or [""])[:2] == ["python", "-m"]],
    ids=lambda v: v.split("/")[-1] if isinstance(v, str) else
    v.get("name", "c"))
def test_manifest_flags_exist(path, container):
    mod = container["command"][2]
    help_text = subprocess.run(
        [sys.executable, "-m", mod, "--help"],
        capture_output=True, text=True, timeout=60).stdout
    assert help_text, f"{mod} --help produced nothing"
    for arg in container.get("args") or []:
        m = re.match(r"(--[a-z0-9-]+)", str(arg))
        if m:
            assert m.group(1) in help_text, \
                f"{path}: {mod} does not accept {m.group(1)}"


def test_chart_values_references_resolve():
    """Every `.Values.x.y` referenced in chart templates exists in
    values.yaml (helm is not installed here; this catches the common
    rename drift statically)."""
    chart = os.path.join(REPO, "deploy", "chart", "vgpu-manager")
    values = yaml.safe_load(open(os.path.join(chart, "values.yaml")))
    refs = set()
    for tpl in glob.glob(os.path.join(chart, "templates", "*")):
        for m in re.finditer(r"\.Values\.([A-Za-z0-9_.]+)",
                             open(tpl).read()):
            refs.add(m.group(1))
    assert refs, "no .Values references found — chart gutted?"
    for ref in sorted(refs):
        node = values
        for part in ref.split("."):
            assert isinstance(node, dict) and part in node, \
                f".Values.{ref} not present in values.yaml"
            node = node[part]


def test_rbac_covers_every_rest_resource():
    """Every API (group, resource) the REST client can touch is
    granted by at least one ClusterRole in deploy/rbac.yaml — catches
    'client grew a new resource, rbac never updated' drift."""
    src = open(os.path.join(
        REPO, "vgpu_manager_amd", "client", "kube.py")).read()
    # join implicitly-concatenated string literals so split paths
    # ("…{namespace}"\n  "/resourceclaimtemplates") regex as one
    src = re.sub(r'"\s*\n\s*f?"', "", src)
    used = set()
    for m in re.finditer(
            r'/api/v1/(?:namespaces/\{[^}]*\}/)?([a-z]+)', src):
        used.add(("", m.group(1)))
    for m in re.finditer(
            r'/apis/([a-z0-9.]+)/v[0-9a-z]+/'
            r'(?:namespaces/\{[^}]*\}/)?([a-z]+)', src):
        used.add((m.group(1), m.group(2)))
    for m in re.finditer(r'/pods/\{[^}]*\}/([a-z]+)', src):
        used.add(("", "pods/" + m.group(1)))
    used.discard(("", "namespaces"))  # path component, not a verb target
    assert ("resource.k8s.io", "resourceclaimtemplates") in used

    granted = set()
    for doc in yaml.safe_load_all(
            open(os.path.join(REPO, "deploy", "rbac.yaml"))):
        if not doc or doc.get("kind") != "ClusterRole":
            continue
        for rule in doc.get("rules") or []:
            for g in rule.get("apiGroups") or []:
                for r in rule.get("resources") or []:
                    granted.add((g, r))
    missing = used - granted
    assert not missing, f"REST resources with no RBAC grant: {missing}"


def test_every_shim_env_is_documented():
    """Every env var the C shim reads (getenv/vgpu_getenv/env_* call
    sites) appears in docs/api_reference.md — the doc uses
    `VGPU_X_{A,B}` brace shorthand, so expand it before comparing."""
    src_dir = os.path.join(REPO, "library", "src")
    used = set()
    for f in glob.glob(os.path.join(src_dir, "*.c")):
        s = open(f).read()
        for m in re.finditer(
                r'(?:vgpu_getenv|getenv|env_(?:int|bool|str|u64|size))'
                r'\(\s*"([A-Z][A-Z0-9_]+)"', s):
            used.add(m.group(1))
        for m in re.finditer(r'"((?:VGPU|CUDA)_[A-Z_]+)_%d"', s):
            used.add(m.group(1) + "_<i>")
    assert len(used) > 20, used  # sanity: extraction still works

    doc = re.sub(r"\s+", "", open(os.path.join(
        REPO, "docs", "api_reference.md")).read())
    # expand {A,B,C} groups (possibly several per token) into full names
    names = set()
    for tok in re.findall(r"[A-Z][A-Z0-9_{},<>i]*", doc):
        variants = [""]
        for part in re.split(r"(\{[A-Z0-9_,<>i]+\})", tok):
            if part.startswith("{"):
                opts = part[1:-1].split(",")
                variants = [v + o for v in variants for o in opts]
            else:
                variants = [v + part for v in variants]
        names.update(variants)
    blob = doc  # plain occurrences (incl. inside longer words)
    missing = {e for e in used
               if e not in names and e not in blob}
    assert not missing, f"shim envs missing from api_reference: {missing}"


def test_webhook_config_paths_are_served():
    """Every webhook path registered in deploy/webhook.yaml must be an
    actual route of the admission app (and the reverse: every POST
    route should be reachable from some webhook configuration)."""
    from vgpu_manager_amd.webhook.admission import create_app

    text = open(os.path.join(REPO, "deploy", "webhook.yaml")).read()
    cfg_paths = set(re.findall(r"path:\s*(/webhook/[a-z-]+)", text))
    assert cfg_paths, "no webhook paths found in deploy/webhook.yaml"
    app = create_app()
    served = {r.path for r in app.routes if r.path.startswith("/webhook")}
    assert cfg_paths <= served, \
        f"configured but unserved: {cfg_paths - served}"
    # unreferenced POST routes are fine only if deliberate; today the
    # one exception is validate-volcanojob (volcano validates via its
    # own admission by default)
    unref = served - cfg_paths
    assert unref <= {"/webhook/validate-volcanojob"}, unref
