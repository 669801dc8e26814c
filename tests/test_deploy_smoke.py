"""Whole-product smoke: deploy.py + CLI + manifests as real processes."""

import os
import socket
import subprocess
import sys
import time

import httpx
import pytest

pytestmark = pytest.mark.timeout(180)

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def test_deploy_cli_end_to_end(tmp_path):
    port = free_port()
    env = dict(os.environ, PYTHONPATH=ROOT, FMA_FAKE_GPU="1",
               FMA_GPU_MODE="naive", FMA_ACCELERATORS="GPU-0")
    proc = subprocess.Popen(
        [sys.executable, "-m", "fma_amd.deploy", "--node-name", "node-1",
         "--store-port", str(port), "--metrics-port", str(free_port()),
         "--gpus", "2"],
        env=env, cwd=ROOT, stdout=open(tmp_path / "deploy.log", "wb"),
        stderr=subprocess.STDOUT, start_new_session=True)
    base = f"http://127.0.0.1:{port}"
    try:
        deadline = time.time() + 60
        up = False
        while time.time() < deadline:
            try:
                if httpx.get(base + "/healthz", timeout=2).status_code == 200:
                    up = True
                    break
            except httpx.HTTPError:
                time.sleep(0.3)
        assert up, open(tmp_path / "deploy.log").read().decode("utf-8",
                                                               "replace")

        def cli(*args):
            return subprocess.run(
                [sys.executable, "-m", "fma_amd.cli", "--store-url", base,
                 *args],
                env=env, cwd=ROOT, capture_output=True, text=True, timeout=60)

        r = cli("apply", "-f", "manifests/example.yaml")
        assert r.returncode == 0, r.stderr
        assert "created" in r.stdout

        # requester becomes Ready through the whole stack
        deadline = time.time() + 90
        ready = False
        while time.time() < deadline:
            r = cli("get", "pod", "my-model-request", "-o", "json")
            if r.returncode == 0 and '"type": "Ready", "status": "True"' in \
                    r.stdout.replace("'", '"'):
                ready = True
                break
            if r.returncode == 0 and '"status": "True"' in r.stdout and \
                    '"Ready"' in r.stdout:
                ready = True
                break
            time.sleep(0.5)
        assert ready, cli("get", "pods").stdout + \
            open(tmp_path / "deploy.log").read().decode("utf-8", "replace")[-2000:]

        r = cli("get", "pods")
        assert "my-model-request" in r.stdout
        r = cli("delete", "pod", "my-model-request")
        assert r.returncode == 0
    finally:
        import signal
        try:
            os.killpg(proc.pid, signal.SIGTERM)
        except ProcessLookupError:
            pass
        proc.wait(timeout=15)


def test_dockerfiles_reference_valid_modules():
    """Every container CMD must point at an importable module, and the
    4-image strategy (controller/requester/launcher-gpu/launcher-cpu)
    exists (reference dockerfiles/* builds 4 images incl. a CPU launcher
    for GPU-less e2e)."""
    import importlib
    import os
    import re

    root = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "deploy", "docker")
    files = sorted(os.listdir(root))
    assert files == ["Dockerfile", "Dockerfile.controller",
                     "Dockerfile.launcher.cpu", "Dockerfile.requester"]
    for fn in files:
        text = open(os.path.join(root, fn)).read()
        for mod in re.findall(r'"-m", "([\w.]+)"', text):
            importlib.import_module(mod)
        assert "PYTHONPATH=/app" in text


def test_ci_workflow_lists_real_paths():
    import os

    import yaml

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    wf = yaml.safe_load(open(os.path.join(
        root, ".github", "workflows", "ci.yml")))
    jobs = wf["jobs"]
    assert {"cpu-tests", "kube-e2e", "hip-build", "images"} <= set(jobs)
    for df in jobs["images"]["strategy"]["matrix"]["dockerfile"]:
        assert os.path.exists(os.path.join(root, df)), df


def _render_helm_subset(chart_dir, values):
    """Minimal renderer for the template subset the chart uses:
    {{ .Values.a.b }}, {{- if .Values.a.b }} / {{- end }},
    {{ .Files.Get "path" }}. Enough to validate the shipped chart
    renders to parseable Kubernetes YAML without helm in the image."""
    import os
    import re

    def lookup(path):
        cur = values
        for part in path.split(".")[2:]:  # strip leading .Values
            cur = cur[part]
        return cur

    rendered = {}
    tdir = os.path.join(chart_dir, "templates")
    for fn in sorted(os.listdir(tdir)):
        out_lines = []
        stack = [True]
        for line in open(os.path.join(tdir, fn)):
            m = re.match(r"\s*\{\{-? if (\.Values[.\w]+) \}\}", line)
            if m:
                stack.append(stack[-1] and bool(lookup(m.group(1))))
                continue
            if re.match(r"\s*\{\{-? end \}\}", line):
                stack.pop()
                continue
            if not stack[-1]:
                continue
            line = re.sub(
                r"\{\{ \.Files\.Get \"([^\"]+)\" \}\}",
                lambda m: open(os.path.join(chart_dir, m.group(1))).read(),
                line)
            line = re.sub(r"\{\{ (\.Values[.\w]+) \}\}",
                          lambda m: str(lookup(m.group(1))), line)
            out_lines.append(line)
        rendered[fn] = "".join(out_lines)
    return rendered


def test_helm_chart_renders_to_valid_kubernetes_yaml():
    import os

    import yaml

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    chart = os.path.join(root, "deploy", "charts", "fma-amd")
    values = yaml.safe_load(open(os.path.join(chart, "values.yaml")))
    assert yaml.safe_load(open(os.path.join(chart, "Chart.yaml")))["name"] \
        == "fma-amd"

    rendered = _render_helm_subset(chart, values)
    kinds = []
    for fn, text in rendered.items():
        for doc in yaml.safe_load_all(text):
            if doc:
                kinds.append(doc["kind"])
                assert "metadata" in doc, fn
    assert "Deployment" in kinds and kinds.count("Deployment") == 2
    assert "ServiceAccount" in kinds and "ClusterRole" in kinds
    assert kinds.count("ValidatingAdmissionPolicy") == 2
    assert kinds.count("ValidatingAdmissionPolicyBinding") == 2

    # the ServiceAccount name must satisfy the VAP exemption pattern
    import re as _re
    sa = next(d for d in yaml.safe_load_all(rendered["rbac.yaml"])
              if d and d["kind"] == "ServiceAccount")
    username = (f"system:serviceaccount:{values['namespace']}:"
                f"{sa['metadata']['name']}")
    assert _re.match(
        r"^system:serviceaccount:[^:]+:[^:]*-fma-controllers$", username)

    # toggles prune their sections
    values2 = dict(values, admissionPolicies={"enabled": False})
    rendered2 = _render_helm_subset(chart, values2)
    assert not any(d for d in yaml.safe_load_all(
        rendered2["admission-policies.yaml"]) if d)
