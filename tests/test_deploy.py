"""Deployment material sanity: YAML validity, setup-ca.sh output."""

import glob
import os
import subprocess

import yaml

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


class TestManifests:
    def test_all_yaml_parse(self):
        manifests = glob.glob(os.path.join(REPO_ROOT, "deploy", "**", "*.yaml"),
                              recursive=True)
        assert manifests, "no manifests found"
        for path in manifests:
            with open(path) as f:
                docs = list(yaml.safe_load_all(f))
            assert docs, path
            for doc in docs:
                assert "kind" in doc, path
                assert "apiVersion" in doc, path

    def test_daemonset_wiring(self):
        path = os.path.join(REPO_ROOT, "deploy", "kubernetes", "malloc",
                            "malloc-daemonset.yaml")
        with open(path) as f:
            ds = yaml.safe_load(f)
        spec = ds["spec"]["template"]["spec"]
        assert spec["nodeSelector"] == {"amd.com/oim": "1"}
        names = [c["name"] for c in spec["containers"]]
        assert "oim-csi-driver" in names
        assert "node-driver-registrar" in names
        driver = next(c for c in spec["containers"]
                      if c["name"] == "oim-csi-driver")
        assert any("--oim-registry-address" in a for a in driver["args"])


class TestSetupCa:
    def test_generates_keypairs_and_secret(self, tmp_path):
        script = os.path.join(REPO_ROOT, "deploy", "setup-ca.sh")
        out = tmp_path / "ca"
        proc = subprocess.run(
            ["bash", script, str(out), "component.registry", "user.admin"],
            capture_output=True, text=True, timeout=120)
        assert proc.returncode == 0, proc.stderr
        assert (out / "ca.crt").exists()
        assert (out / "component.registry.key").exists()
        assert (out / "user.admin.crt").exists()
        with open(out / "secret.yaml") as f:
            secret = yaml.safe_load(f)
        assert secret["kind"] == "Secret"
        assert "ca.crt" in secret["data"]
