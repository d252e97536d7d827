"""Black-box CLI e2e: drive the real `bin/kuke` binary as a subprocess
against a per-test run path (the reference's e2e style — SURVEY.md §4
tier 3: CLI output, on-disk metadata and process state verified together).
Uses --local (in-process verbs) so no daemon needs to be running."""
import os
import subprocess
import time
from pathlib import Path

import yaml

REPO = Path(__file__).resolve().parent.parent
KUKE = str(REPO / "bin" / "kuke")

CELL_YAML = """
apiVersion: v1beta1
kind: Cell
metadata: {name: cliy}
spec:
  realmId: default
  spaceId: default
  stackId: default
  containers:
    - id: main
      image: busybox
      command: sleep
      args: ["30"]
"""


def kuke(run, *args, check=True, env_extra=None):
    env = dict(os.environ)
    env["KUKE_CONFIG"] = str(Path(run) / "nonexistent-kuke.yaml")
    if env_extra:
        env.update(env_extra)
    r = subprocess.run([KUKE, "--run-path", run, "--local", *args],
                       capture_output=True, text=True, timeout=60, env=env)
    if check:
        assert r.returncode == 0, (args, r.stdout, r.stderr)
    return r


def test_cli_full_lifecycle(tmp_path):
    run = str(tmp_path / "run")
    kuke(run, "init")
    assert (Path(run) / "data" / "default" / "metadata.json").exists()

    spec = tmp_path / "cell.yaml"
    spec.write_text(CELL_YAML)
    out = kuke(run, "apply", "-f", str(spec)).stdout
    assert "created" in out

    out = kuke(run, "get", "cells").stdout
    assert "cliy" in out

    kuke(run, "start", "cliy")
    doc = yaml.safe_load(kuke(run, "get", "cell", "cliy", "-o",
                              "yaml").stdout)
    assert doc["status"]["state"] == "Ready"
    pid = doc["status"]["containers"][0]["pid"]
    assert pid > 0

    top = kuke(run, "top", "cliy").stdout
    assert "TOTAL" in top and "main" in top

    out = kuke(run, "status", check=False)
    assert "state tree" in out.stdout

    kuke(run, "stop", "cliy")
    deadline = time.monotonic() + 5
    while time.monotonic() < deadline:
        try:
            os.kill(pid, 0)
            time.sleep(0.05)
        except ProcessLookupError:
            break
    doc = yaml.safe_load(kuke(run, "get", "cell", "cliy", "-o",
                              "yaml").stdout)
    assert doc["status"]["state"] == "Stopped"

    kuke(run, "delete", "cell", "cliy")
    r = kuke(run, "get", "cell", "cliy", check=False)
    assert r.returncode != 0


def test_cli_apply_validation_error(tmp_path):
    run = str(tmp_path / "run")
    kuke(run, "init")
    bad = tmp_path / "bad.yaml"
    bad.write_text("apiVersion: v1beta1\nkind: Cell\n"
                   "metadata: {name: BAD NAME}\n"
                   "spec: {realmId: default, spaceId: default, "
                   "stackId: default}\n")
    r = kuke(run, "apply", "-f", str(bad), check=False)
    assert r.returncode != 0 or "failed" in r.stdout


def test_cli_against_live_daemon(tmp_path):
    """kuke over the unix socket to a live kukeond (not --local)."""
    import uuid

    from kukeon_amd.controller.core import Controller
    from kukeon_amd.daemon.server import Server

    run = str(tmp_path / "run")
    sock = f"/tmp/kukecli-{uuid.uuid4().hex[:8]}.sock"
    ctl = Controller(run, gpu_devices=[])
    ctl.bootstrap()
    srv = Server(ctl, sock, reconcile_interval=0)
    srv.start()
    try:
        env = dict(os.environ)
        env["KUKE_CONFIG"] = str(tmp_path / "none.yaml")

        def kd(*args, check=True):
            r = subprocess.run([KUKE, "--run-path", run, "--socket", sock,
                                *args], capture_output=True, text=True,
                               timeout=60, env=env)
            if check:
                assert r.returncode == 0, (args, r.stdout, r.stderr)
            return r

        spec = tmp_path / "cell.yaml"
        spec.write_text(CELL_YAML)
        assert "created" in kd("apply", "-f", str(spec)).stdout
        kd("start", "cliy")
        assert "Ready" in kd("get", "cell", "cliy", "-o", "yaml").stdout
        st = kd("status")
        assert "daemon rpc" in st.stdout and "[  ok]" in st.stdout
        kd("kill", "cliy")
        kd("delete", "cell", "cliy")
    finally:
        srv.stop()


def test_cli_session_lifecycle(tmp_path):
    run = str(tmp_path / "run")
    kuke(run, "init")
    out = kuke(run, "session", "create", "agent-7", "--gpus", "0",
               "--wall-clock", "30m", "--task", "demo").stdout
    assert "Running" in out
    # dedicated stack named after the session was created
    assert (Path(run) / "data" / "default" / "default" / "agent-7").exists()
    out = kuke(run, "get", "session", "agent-7", "-o", "yaml").stdout
    assert "Running" in out
    out = kuke(run, "session", "close", "agent-7").stdout
    assert "Completed" in out


def test_cli_run_divergence_warning(tmp_path):
    run = str(tmp_path / "run")
    kuke(run, "init")
    spec = tmp_path / "cell.yaml"
    spec.write_text(CELL_YAML)
    kuke(run, "run", "-f", str(spec), "--no-attach")
    # same file again: no divergence warning
    r = kuke(run, "run", "-f", str(spec), "--no-attach")
    assert "diverging" not in r.stderr
    # changed args: reuse with a warning
    spec.write_text(CELL_YAML.replace('["30"]', '["60"]'))
    r = kuke(run, "run", "-f", str(spec), "--no-attach")
    assert "diverging spec" in r.stderr and "args" in r.stderr
    kuke(run, "kill", "cliy")


def test_cli_doctor(tmp_path):
    run = str(tmp_path / "run")
    r = kuke(run, "doctor")
    out = r.stdout
    for field in ("cgroups", "cgroup:memory", "amdgpu", "kfd", "iptables",
                  "git", "run path", "hip extension"):
        assert field in out, out


def test_get_api_version_downgrade(tmp_path):
    """kuke get --api-version v1alpha1 exports the lossy downgrade
    (scope keys renamed, dropped beta-only fields reported)."""
    import yaml as yamlmod
    from click.testing import CliRunner
    from kukeon_amd.cli.main import cli

    runner = CliRunner()
    rp = str(tmp_path / "run")
    r = runner.invoke(cli, ["--run-path", rp, "--local", "init"])
    assert r.exit_code == 0, r.output
    cell = {
        "apiVersion": "v1beta1", "kind": "Cell",
        "metadata": {"name": "alpha-export"},
        "spec": {"realmId": "default", "spaceId": "default",
                 "stackId": "default",
                 "containers": [{"id": "main", "command": "sleep",
                                 "args": ["5"],
                                 "repos": [{"url": "https://x/y.git"}]}]},
    }
    f = tmp_path / "cell.yaml"
    f.write_text(yamlmod.safe_dump(cell))
    r = runner.invoke(cli, ["--run-path", rp, "--local", "apply", "-f",
                            str(f)])
    assert r.exit_code == 0, r.output
    r = runner.invoke(cli, ["--run-path", rp, "--local", "get", "cell",
                            "alpha-export", "-o", "yaml",
                            "--api-version", "v1alpha1"])
    assert r.exit_code == 0, r.output
    docs = list(yamlmod.safe_load_all(
        "\n".join(l for l in r.output.splitlines()
                  if not l.startswith("warn:"))))
    assert docs[0]["apiVersion"] == "v1alpha1"
    assert docs[0]["spec"]["realm"] == "default"  # scope key renamed
    assert "realmId" not in docs[0]["spec"]
    assert "dropped in v1alpha1" in r.output  # repos is beta-only


def test_daemon_recreate_reprovisions_modelhub(tmp_path):
    from click.testing import CliRunner
    from kukeon_amd.cli.main import cli

    runner = CliRunner()
    rp = str(tmp_path / "run")
    assert runner.invoke(cli, ["--run-path", rp, "--local",
                               "init"]).exit_code == 0
    r = runner.invoke(cli, ["--run-path", rp, "--local", "daemon",
                            "recreate", "--gpus", "0"])
    assert r.exit_code == 0, r.output
    assert "modelhub cell recreated" in r.output
    # idempotent: a second recreate deletes and re-provisions
    r2 = runner.invoke(cli, ["--run-path", rp, "--local", "daemon",
                             "recreate", "--gpus", "0"])
    assert r2.exit_code == 0, r2.output


def test_version_and_autocomplete_smoke():
    from click.testing import CliRunner
    from kukeon_amd.cli.main import cli

    runner = CliRunner()
    r = runner.invoke(cli, ["version"])
    assert r.exit_code == 0 and "kuke" in r.output
    r = runner.invoke(cli, ["autocomplete"])
    assert r.exit_code == 0 and "complete" in r.output.lower()


def test_client_configuration_defaults(tmp_path, monkeypatch):
    """~/.kuke/kuke.yaml (ClientConfiguration) supplies default scope:
    a cell created in a non-default realm/space is found by `kuke get`
    without --realm/--space flags."""
    import yaml as yamlmod
    from click.testing import CliRunner
    from kukeon_amd.cli.main import cli

    runner = CliRunner()
    rp = str(tmp_path / "run")
    assert runner.invoke(cli, ["--run-path", rp, "--local",
                               "init"]).exit_code == 0
    docs = [
        {"apiVersion": "v1beta1", "kind": "Realm",
         "metadata": {"name": "prod"}},
        {"apiVersion": "v1beta1", "kind": "Space",
         "metadata": {"name": "web"}, "spec": {"realmId": "prod"}},
        {"apiVersion": "v1beta1", "kind": "Stack",
         "metadata": {"name": "app"},
         "spec": {"realmId": "prod", "spaceId": "web"}},
        {"apiVersion": "v1beta1", "kind": "Cell",
         "metadata": {"name": "scoped"},
         "spec": {"realmId": "prod", "spaceId": "web", "stackId": "app",
                  "containers": [{"id": "main", "command": "sleep",
                                  "args": ["5"]}]}},
    ]
    f = tmp_path / "m.yaml"
    f.write_text(yamlmod.safe_dump_all(docs))
    assert runner.invoke(cli, ["--run-path", rp, "--local", "apply",
                               "-f", str(f)]).exit_code == 0
    cfgp = tmp_path / "kuke.yaml"
    cfgp.write_text(yamlmod.safe_dump({
        "apiVersion": "v1beta1", "kind": "ClientConfiguration",
        "spec": {"defaultRealm": "prod", "defaultSpace": "web",
                 "defaultStack": "app"}}))
    monkeypatch.setenv("KUKE_CONFIG", str(cfgp))
    r = runner.invoke(cli, ["--run-path", rp, "--local", "get", "cell",
                            "scoped"])
    assert r.exit_code == 0 and "scoped" in r.output, r.output
    # without the config the default scope misses it
    monkeypatch.setenv("KUKE_CONFIG", str(tmp_path / "none.yaml"))
    r2 = runner.invoke(cli, ["--run-path", rp, "--local", "get", "cell",
                             "scoped"])
    assert r2.exit_code != 0


def test_uninstall_removes_state(tmp_path):
    """kuke uninstall: cascade-deletes realms and removes the run tree
    (reference internal/controller/uninstall.go)."""
    from click.testing import CliRunner
    from kukeon_amd.cli.main import cli

    runner = CliRunner()
    rp = str(tmp_path / "run")
    assert runner.invoke(cli, ["--run-path", rp, "--local",
                               "init"]).exit_code == 0
    assert (Path(rp) / "data").is_dir()
    r = runner.invoke(cli, ["--run-path", rp, "--local", "uninstall",
                            "--yes"])
    assert r.exit_code == 0, r.output
    assert not (Path(rp) / "data").exists()
