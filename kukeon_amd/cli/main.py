"""kuke — the CLI. `python -m kukeon_amd.cli` or the bin/kuke wrapper.

Verb surface (reference cmd/kuke parity): init, apply, run, attach, get,
create, delete, purge, start, stop, kill, restart, log, status, doctor,
session, daemon, version. Talks JSON-RPC to kukeond over the unix socket;
`--local` (or a missing daemon socket during init) promotes verbs in-process
over a Controller, like the reference's image path.
"""
from __future__ import annotations

import json
import os
import sys
import time
from pathlib import Path

import click
import yaml

from kukeon_amd.api import errors
from kukeon_amd.api import v1beta1 as api
from kukeon_amd.api.client import LocalClient, UnixClient

DEFAULT_RUN_PATH = os.environ.get("KUKEON_RUN_PATH", "/run/kukeon")


def load_client_config() -> dict:
    p = Path(os.environ.get("KUKE_CONFIG",
                            os.path.expanduser("~/.kuke/kuke.yaml")))
    if p.exists():
        try:
            doc = yaml.safe_load(p.read_text()) or {}
            return doc.get("spec", doc)
        except Exception:
            return {}
    return {}


class Ctx:
    def __init__(self, socket_path, run_path, local, realm, space, stack):
        cfg = load_client_config()
        self.socket_path = socket_path or cfg.get("socket") or os.environ.get(
            "KUKEOND_SOCKET") or str(Path(run_path) / "kukeond.sock")
        self.run_path = run_path
        self.local = local
        self.realm = realm or cfg.get("defaultRealm") or "default"
        self.space = space or cfg.get("defaultSpace") or "default"
        self.stack = stack or cfg.get("defaultStack") or "default"
        self._client = None

    @property
    def client(self):
        if self._client is None:
            if self.local or not os.path.exists(self.socket_path):
                self._client = LocalClient(self._controller())
            else:
                self._client = UnixClient(self.socket_path)
        return self._client

    def _controller(self, server_cfg=None):
        from kukeon_amd.controller.core import Controller
        from kukeon_amd.runtime.cgroup import CgroupManager
        from kukeon_amd.netpolicy import make_enforcer
        ctl = Controller(self.run_path, cgroups=CgroupManager(),
                         enforcer=make_enforcer(), server_config=server_cfg)
        return ctl


pass_ctx = click.make_pass_decorator(Ctx)


@click.group()
@click.option("--socket", "socket_path", default=None,
              help="kukeond unix socket path")
@click.option("--run-path", default=DEFAULT_RUN_PATH, show_default=True)
@click.option("--local", is_flag=True, help="run verbs in-process")
@click.option("--realm", default=None)
@click.option("--space", default=None)
@click.option("--stack", default=None)
@click.pass_context
def cli(ctx, socket_path, run_path, local, realm, space, stack):
    ctx.obj = Ctx(socket_path, run_path, local, realm, space, stack)


def _die(e: Exception):
    click.echo(f"error: {e}", err=True)
    sys.exit(1)


@cli.command()
@pass_ctx
def init(ctx):
    """Bootstrap the run path, hierarchies, cgroups and the kukeon
    system group (reference kuke init: sysuser.EnsureUserGroup +
    ownership fixups — non-root hosts degrade to single-user)."""
    ctl = ctx._controller()
    ctl.bootstrap()
    from kukeon_amd.daemon.server import verify_or_write_instance
    verify_or_write_instance(Path(ctx.run_path))
    from kukeon_amd.runtime import sysuser
    gid = sysuser.ensure_group()
    if gid is not None:
        changed = sysuser.chown_tree(Path(ctx.run_path), gid)
        click.echo(f"group kukeon (gid {gid}): {changed} entries owned")
    click.echo(f"initialized kukeon at {ctx.run_path} "
               f"(cgroups: {ctl.cgroups.mode}, "
               f"gpus: {ctl.gpus.devices or 'none'})")


@cli.command()
@click.option("-f", "--file", "files", multiple=True, required=True,
              type=click.Path(exists=True, allow_dash=True))
@click.option("--team", default="")
@pass_ctx
def apply(ctx, files, team):
    """Apply declarative manifests (multi-doc YAML)."""
    for fp in files:
        text = (sys.stdin.read() if fp == "-" else
                Path(fp).read_text())
        try:
            results = ctx.client.ApplyDocuments(yaml=text, team=team)
        except errors.KukeonError as e:
            _die(e)
        rc = 0
        for r in results:
            line = f"{r['kind'].lower()}/{r['name']} {r['action']}"
            if r["error"]:
                line += f": {r['error']}"
                rc = 1
            click.echo(line)
        if rc:
            sys.exit(rc)


@cli.command()
@click.option("-f", "--file", "file_", type=click.Path(exists=True))
@click.option("-b", "--blueprint", default=None)
@click.option("-c", "--config", "config_", default=None)
@click.option("-p", "--param", "params", multiple=True,
              help="KEY=VALUE blueprint params")
@click.option("--env", "env_", multiple=True, help="KEY=VALUE runtime env")
@click.option("--name", default=None)
@click.option("--rm", "auto_delete", is_flag=True)
@click.option("--attach/--no-attach", default=True)
@pass_ctx
def run(ctx, file_, blueprint, config_, params, env_, name, auto_delete,
        attach):
    """Create (or reuse) a cell and attach to it."""
    kv = dict(p.split("=", 1) for p in params)
    try:
        if file_:
            raw = yaml.safe_load(Path(file_).read_text())
            doc = api.CellDoc.from_dict(raw)
            doc.spec.auto_delete = doc.spec.auto_delete or auto_delete
            doc.spec.realm_id = doc.spec.realm_id or ctx.realm
            doc.spec.space_id = doc.spec.space_id or ctx.space
            doc.spec.stack_id = doc.spec.stack_id or ctx.stack
            try:
                cell = ctx.client.GetCell(
                    realm=doc.spec.realm_id, space=doc.spec.space_id,
                    stack=doc.spec.stack_id, name=doc.metadata.name)
                # divergence warning (reference run.go create-or-attach):
                # the file's spec differs from the live cell
                from kukeon_amd.controller import diff as diffmod
                d = diffmod.diff_cell(doc, api.CellDoc.from_dict(cell))
                if d.change_type is not diffmod.ChangeType.NONE:
                    click.echo(
                        "warning: cell exists with a diverging spec "
                        f"({', '.join(d.paths)}) — reusing the live cell; "
                        "`kuke apply -f` to update it", err=True)
                click.echo(f"cell {doc.metadata.name} exists — starting")
            except errors.NotFound:
                ctx.client.CreateCell(doc=doc.to_dict(),
                                      runtimeEnv=list(env_))
            cell = ctx.client.StartCell(
                realm=doc.spec.realm_id, space=doc.spec.space_id,
                stack=doc.spec.stack_id, name=doc.metadata.name)
        elif blueprint:
            cell = ctx.client.RunFromBlueprint(
                realm=ctx.realm, space=ctx.space, stack=ctx.stack,
                blueprint=blueprint, params=kv, env=list(env_), name=name)
        elif config_:
            cell = ctx.client.RunFromConfig(
                realm=ctx.realm, space=ctx.space, stack=ctx.stack,
                config=config_, params=kv, name=name)
        else:
            _die(errors.InvalidArgument("one of -f / -b / -c is required"))
    except errors.KukeonError as e:
        _die(e)
    cname = cell["metadata"]["name"]
    click.echo(f"cell {cname}: {cell['status']['state']}")
    has_attachable = any(c.get("attachable")
                         for c in cell["spec"]["containers"])
    if attach and has_attachable:
        _attach_cell(ctx, cell["spec"]["realmId"], cell["spec"]["spaceId"],
                     cell["spec"]["stackId"], cname)


def _attach_cell(ctx, realm, space, stack, name, timeout=10.0):
    from kukeon_amd.tty import attach as attach_mod
    deadline = time.monotonic() + timeout
    path = None
    while time.monotonic() < deadline:
        try:
            path = ctx.client.AttachContainer(realm=realm, space=space,
                                              stack=stack, name=name)
            path = path["hostSocketPath"]
            if attach_mod.ping(path):
                break
        except errors.KukeonError:
            pass
        time.sleep(0.2)
    else:
        _die(errors.AttachPingTimeout(f"cell {name} tty never became ready"))
    click.echo("(attached — detach: Ctrl-] Ctrl-])", err=True)
    rc = attach_mod.attach(path)
    click.echo("", err=True)
    sys.exit(rc if rc != 2 else 0)


@cli.command()
@click.argument("name")
@pass_ctx
def attach(ctx, name):
    """Attach to a running cell's tty."""
    _attach_cell(ctx, ctx.realm, ctx.space, ctx.stack, name)


KIND_ALIASES = {
    "realm": "Realm", "realms": "Realm", "space": "Space", "spaces": "Space",
    "stack": "Stack", "stacks": "Stack", "cell": "Cell", "cells": "Cell",
    "session": "Session", "sessions": "Session", "secret": "Secret",
    "secrets": "Secret", "blueprint": "CellBlueprint",
    "blueprints": "CellBlueprint", "config": "CellConfig",
    "configs": "CellConfig", "volume": "Volume", "volumes": "Volume",
}


@cli.command()
@click.argument("kind")
@click.argument("name", required=False)
@click.option("-o", "--output", default="table",
              type=click.Choice(["table", "yaml", "json"]))
@click.option("--api-version", "api_version", default=None,
              type=click.Choice(["v1alpha1", "v1beta1"]),
              help="export documents at a wire version (v1alpha1 is a "
                   "lossy downgrade; dropped fields are reported on "
                   "stderr)")
@pass_ctx
def get(ctx, kind, name, output, api_version):
    """Get resources: kuke get cells | kuke get cell NAME -o yaml."""
    k = KIND_ALIASES.get(kind.lower())
    if k is None:
        _die(errors.InvalidArgument(f"unknown kind {kind}"))
    c, r, s, st = ctx.client, ctx.realm, ctx.space, ctx.stack
    try:
        if k == "Realm":
            docs = [c.GetRealm(name=name)] if name else c.ListRealms()
        elif k == "Space":
            docs = [c.GetSpace(realm=r, name=name)] if name else \
                c.ListSpaces(realm=r)
        elif k == "Stack":
            docs = [c.GetStack(realm=r, space=s, name=name)] if name else \
                c.ListStacks(realm=r, space=s)
        elif k == "Cell":
            docs = [c.GetCell(realm=r, space=s, stack=st, name=name)] \
                if name else c.ListCells(realm=r, space=s, stack=st)
        elif k == "Session":
            if name:
                try:
                    docs = [c.GetSession(realm=r, space=s, stack=st,
                                         name=name)]
                except errors.NotFound:
                    # dedicated-stack convention: the session's stack is
                    # named after it
                    docs = [c.GetSession(realm=r, space=s, stack=name,
                                         name=name)]
            else:
                docs = c.ListSessions()
        elif k == "Secret":
            docs = ([c.GetSecret(realm=r, space=s, name=name)] if name else
                    [{"metadata": {"name": n}, "kind": "Secret",
                      "status": {"state": "-"}}
                     for n in c.ListSecrets(realm=r, space=s)])
        elif k == "CellBlueprint":
            docs = ([c.GetBlueprint(realm=r, space=s, name=name)] if name
                    else [{"metadata": {"name": n}, "kind": "CellBlueprint",
                           "status": {"state": "-"}}
                          for n in c.ListBlueprints(realm=r, space=s)])
        elif k == "CellConfig":
            docs = ([c.GetConfig(realm=r, space=s, name=name)] if name else
                    [{"metadata": {"name": n}, "kind": "CellConfig",
                      "status": {"state": "-"}}
                     for n in c.ListConfigs(realm=r, space=s)])
        elif k == "Volume":
            docs = [c.GetVolume(realm=r, space=s, name=name)]
        else:
            docs = []
    except errors.KukeonError as e:
        _die(e)
    if api_version and api_version != "v1beta1":
        from kukeon_amd.api import scheme
        wired = []
        for d in docs:
            w, lost = scheme.to_wire(d, api_version)
            wired.append(w)
            for f in lost:
                click.echo(f"warn: {d['metadata']['name']}: dropped in "
                           f"{api_version}: {f}", err=True)
        docs = wired
        if output == "table":
            output = "yaml"  # a downgraded doc only makes sense as a doc
    if output == "yaml":
        click.echo(yaml.safe_dump_all(docs, sort_keys=False).rstrip())
    elif output == "json":
        click.echo(json.dumps(docs, indent=2))
    else:
        click.echo(f"{'NAME':<28} {'KIND':<14} {'STATE':<10}")
        for d in docs:
            click.echo(f"{d['metadata']['name']:<28} "
                       f"{d.get('kind', k):<14} "
                       f"{d.get('status', {}).get('state', '-'):<10}")


def _scope_verb(ctx, verb, kind, name, cascade=False, force=False):
    c, r, s, st = ctx.client, ctx.realm, ctx.space, ctx.stack
    k = KIND_ALIASES.get(kind.lower())
    try:
        if verb == "delete":
            if k == "Realm":
                c.DeleteRealm(name=name, cascade=cascade)
            elif k == "Space":
                c.DeleteSpace(realm=r, name=name, cascade=cascade)
            elif k == "Stack":
                c.DeleteStack(realm=r, space=s, name=name, cascade=cascade)
            elif k == "Cell":
                c.DeleteCell(realm=r, space=s, stack=st, name=name,
                             force=force)
            elif k == "Session":
                try:
                    c.DeleteSession(realm=r, space=s, stack=st, name=name)
                except errors.NotFound:
                    # dedicated-stack convention
                    c.DeleteSession(realm=r, space=s, stack=name, name=name)
            elif k == "Secret":
                c.DeleteSecret(realm=r, space=s, name=name)
            elif k == "CellBlueprint":
                c.DeleteBlueprint(realm=r, space=s, name=name)
            elif k == "CellConfig":
                c.DeleteConfig(realm=r, space=s, name=name)
            elif k == "Volume":
                c.DeleteVolume(realm=r, space=s, name=name)
            else:
                _die(errors.InvalidArgument(f"cannot delete kind {kind}"))
    except errors.KukeonError as e:
        _die(e)
    click.echo(f"{kind}/{name} deleted")


@cli.command()
@click.argument("kind")
@click.argument("name")
@click.option("--cascade", is_flag=True)
@click.option("--force", is_flag=True)
@pass_ctx
def delete(ctx, kind, name, cascade, force):
    """Delete a resource (graceful; --cascade for scopes)."""
    _scope_verb(ctx, "delete", kind, name, cascade, force)


@cli.command()
@click.argument("kind")
@click.argument("name")
@pass_ctx
def purge(ctx, kind, name):
    """Force residual-state removal of a cell."""
    if KIND_ALIASES.get(kind.lower()) != "Cell":
        _die(errors.InvalidArgument("purge supports cells"))
    try:
        ctx.client.PurgeCell(realm=ctx.realm, space=ctx.space,
                             stack=ctx.stack, name=name)
    except errors.KukeonError as e:
        _die(e)
    click.echo(f"cell/{name} purged")


def _lifecycle(ctx, method, name):
    try:
        cell = ctx.client.call(method, realm=ctx.realm, space=ctx.space,
                               stack=ctx.stack, name=name)
        click.echo(f"cell {name}: {cell['status']['state']}")
    except errors.KukeonError as e:
        _die(e)


@cli.command()
@click.argument("name")
@pass_ctx
def start(ctx, name):
    """Start a cell."""
    _lifecycle(ctx, "StartCell", name)


@cli.command()
@click.argument("name")
@pass_ctx
def stop(ctx, name):
    """Gracefully stop a cell (SIGTERM, 10s, SIGKILL)."""
    _lifecycle(ctx, "StopCell", name)


@cli.command()
@click.argument("name")
@pass_ctx
def kill(ctx, name):
    """Kill a cell immediately."""
    _lifecycle(ctx, "KillCell", name)


@cli.command()
@click.argument("name")
@click.option("-c", "--container", default="",
              help="restart only this container of the cell")
@pass_ctx
def restart(ctx, name, container):
    """Restart a cell (or one container of it with -c)."""
    if container:
        try:
            cell = ctx.client.call("RestartContainer", realm=ctx.realm,
                                   space=ctx.space, stack=ctx.stack,
                                   name=name, container=container)
            click.echo(f"cell {name}/{container}: "
                       f"{cell['status']['state']}")
        except errors.KukeonError as e:
            _die(e)
        return
    _lifecycle(ctx, "RestartCell", name)


@cli.command()
@click.argument("name")
@pass_ctx
def top(ctx, name):
    """Live per-container CPU/RSS for a cell (TaskMetrics parity)."""
    try:
        m = ctx.client.CellMetrics(realm=ctx.realm, space=ctx.space,
                                   stack=ctx.stack, name=name)
    except errors.KukeonError as e:
        _die(e)
    click.echo(f"{'CONTAINER':<20} {'PID':>7} {'CPU(s)':>9} "
               f"{'RSS(MB)':>9} {'THREADS':>8}")
    for cname, c in m["containers"].items():
        if not c.get("running"):
            click.echo(f"{cname:<20} {'-':>7} {'-':>9} {'-':>9} {'-':>8}")
            continue
        click.echo(f"{cname:<20} {c['pid']:>7} {c['cpuSeconds']:>9.1f} "
                   f"{c['rssBytes'] / 1e6:>9.1f} {c['threads']:>8}")
    t = m["total"]
    click.echo(f"{'TOTAL':<20} {'':>7} {t['cpuSeconds']:>9.1f} "
               f"{t['rssBytes'] / 1e6:>9.1f} {t['threads']:>8}")


@cli.command()
@click.argument("name")
@click.option("--container", default="")
@click.option("-f", "--follow", is_flag=True)
@click.option("-n", "--lines", default=100)
@pass_ctx
def log(ctx, name, container, follow, lines):
    """Show a cell's capture/log file (path resolved by the daemon; bytes
    never cross the RPC)."""
    try:
        res = ctx.client.LogPath(realm=ctx.realm, space=ctx.space,
                                 stack=ctx.stack, name=name,
                                 container=container)
    except errors.KukeonError as e:
        _die(e)
    path = Path(res["path"])
    if not path.exists():
        _die(errors.NotFound(f"no log at {path}"))
    raw = path.read_bytes()
    rot = Path(str(path) + ".1")   # shim rotation keeps one generation
    if len(raw.splitlines()) < lines and rot.exists():
        raw = rot.read_bytes() + raw
    data = raw.splitlines()[-lines:]
    for line in data:
        click.echo(line.decode("utf-8", "replace"))
    if follow:
        with open(path, "rb") as f:
            f.seek(0, 2)
            try:
                while True:
                    chunk = f.read(65536)
                    if chunk:
                        sys.stdout.buffer.write(chunk)
                        sys.stdout.flush()
                    else:
                        time.sleep(0.25)
            except KeyboardInterrupt:
                pass


@cli.command()
@pass_ctx
def status(ctx):
    """Daemon/host/state checks."""
    checks = []
    sock = Path(ctx.socket_path)
    checks.append(("daemon socket", sock.exists(), str(sock)))
    if sock.exists():
        try:
            st = ctx.client.Status()
            checks.append(("daemon rpc", True, f"pid {st['pid']}"))
            checks.append(("cgroups", st["cgroupMode"] != "none",
                           st["cgroupMode"]))
            g = st["gpus"]
            checks.append(("gpus", True,
                           f"{len(g['devices'])} devices, "
                           f"{len(g['free'])} free"))
        except Exception as e:
            checks.append(("daemon rpc", False, str(e)))
    data = Path(ctx.run_path) / "data"
    checks.append(("state tree", data.is_dir(), str(data)))
    if data.is_dir():
        # per-realm storage stats (reference ctr/client.go:169-181 exposes
        # per-namespace storage from boltdb; here the state tree + per-run
        # image layers are the storage domains)
        def _du(p: Path) -> int:
            total = 0
            for f in p.rglob("*"):
                try:
                    if f.is_file() and not f.is_symlink():
                        total += f.stat().st_size
                except OSError:
                    continue
            return total

        for realm in sorted(d.name for d in data.iterdir() if d.is_dir()):
            checks.append((f"storage:{realm}", True,
                           f"{_du(data / realm) / 1024:.0f} KiB"))
        layers = Path(ctx.run_path) / "layers"
        if layers.is_dir():
            checks.append(("storage:layers", True,
                           f"{_du(layers) / (1 << 20):.1f} MiB"))
    ok = True
    for name, good, detail in checks:
        mark = "ok" if good else "FAIL"
        ok &= good
        click.echo(f"[{mark:>4}] {name:<16} {detail}")
    sys.exit(0 if ok else 1)


@cli.command()
@click.argument("what", default="all")
@pass_ctx
def doctor(ctx, what):
    """Preflight checks with required/optional classification (reference
    cgroupcheck probe-and-classify); exits non-zero when a required
    capability is missing."""
    from kukeon_amd.runtime.cgroup import CgroupManager
    from kukeon_amd.runtime.devices import discover_gpus
    import shutil as sh
    ok = True

    def row(name, good, detail, required=False):
        nonlocal ok
        mark = "ok" if good else ("FAIL" if required else "warn")
        if required and not good:
            ok = False
        click.echo(f"[{mark:>4}] {name:<16} {detail}")

    cg = CgroupManager()
    ctrls = set(cg.available_controllers())
    row("cgroups", cg.mode != "none",
        f"mode={cg.mode} controllers={','.join(sorted(ctrls)) or '-'}")
    for c in ("cpu", "memory", "pids"):
        row(f"cgroup:{c}", c in ctrls or cg.mode == "none",
            "available" if c in ctrls else "missing (limits degrade)")
    gpus = discover_gpus()
    row("amdgpu", True, f"{len(gpus)} device(s) {gpus}")
    row("kfd", os.path.exists("/dev/kfd"),
        "present" if os.path.exists("/dev/kfd") else
        "absent (no GPU compute on this host)")
    row("iptables", bool(sh.which("iptables")),
        "present" if sh.which("iptables") else
        "absent (egress policy degrades to noop)")
    row("git", bool(sh.which("git")),
        "present" if sh.which("git") else
        "absent (container repo setup unavailable)")
    from kukeon_amd.runtime import namespaces as nsmod
    from kukeon_amd.runtime import netlink as nlmod
    from kukeon_amd.runtime import sysuser
    from kukeon_amd.images import overlay_supported
    ns_ok = nsmod.can_unshare(nsmod.CLONE_NEWUTS | nsmod.CLONE_NEWIPC)
    row("namespaces", ns_ok,
        "uts/ipc/mount available" if ns_ok else
        "denied (cells run in host namespaces)")
    net_ok = nsmod.can_unshare(nsmod.CLONE_NEWNET) and nlmod.available()
    row("netns+rtnetlink", net_ok,
        "per-space bridges available" if net_ok else
        "denied (space networks degrade to host net)")
    ovl = overlay_supported()
    row("overlayfs", ovl,
        "image rootfs + RUN builds available" if ovl else
        "absent (COPY-only builds, direct-chroot rootfs)")
    gid = sysuser.lookup_group()
    row("kukeon group", gid is not None,
        f"gid {gid}" if gid is not None else
        "absent (kuke init creates it; single-user until then)")
    run = Path(ctx.run_path)
    writable = os.access(run if run.exists() else run.parent, os.W_OK)
    row("run path", writable, str(run), required=True)
    try:
        import kukeon_amd.ops as ops
        row("hip extension", ops.native_available(),
            "loaded" if ops.native_available() else
            "not built (python -m kukeon_amd.ops.build)")
    except Exception as e:  # noqa: BLE001
        row("hip extension", False, f"error ({e})")
    sys.exit(0 if ok else 1)


@cli.group()
def session():
    """Session lifecycle verbs."""


@session.command("create")
@click.argument("name")
@click.option("--task", default="")
@click.option("--owner", default="")
@click.option("--gpus", default=0)
@click.option("--wall-clock", default="")
@click.option("--idle-timeout", default="")
@click.option("--session-stack", default="",
              help="stack the session owns (default: a dedicated stack "
                   "named after the session, swept on close)")
@pass_ctx
def session_create(ctx, name, task, owner, gpus, wall_clock, idle_timeout,
                   session_stack):
    doc = api.SessionDoc(
        metadata=api.Metadata(name=name),
        spec=api.SessionSpec(
            realm_id=ctx.realm, space_id=ctx.space,
            stack_id=session_stack or name,
            owner=owner, task=task, gpus=gpus,
            lifetime=api.SessionLifetime(wall_clock=wall_clock,
                                         idle_timeout=idle_timeout)
            if (wall_clock or idle_timeout) else None))
    try:
        res = ctx.client.CreateSession(doc=doc.to_dict())
    except errors.KukeonError as e:
        _die(e)
    click.echo(f"session {name}: {res['status']['state']} "
               f"gpus={res['status'].get('gpuIds', [])} "
               f"deadline={res['status'].get('deadline', '-')}")


@session.command("close")
@click.argument("name")
@pass_ctx
@click.option("--session-stack", default="")
def session_close(ctx, name, session_stack):
    try:
        res = ctx.client.CloseSession(realm=ctx.realm, space=ctx.space,
                                      stack=session_stack or name,
                                      name=name)
    except errors.KukeonError as e:
        _die(e)
    click.echo(f"session {name}: {res['status']['state']}")


@cli.group()
def team():
    """Team distribution (kuke team init)."""


@team.command("init")
@click.option("-f", "--file", "file_", default="kuketeam.yaml",
              type=click.Path(exists=True))
@click.option("--teams-root", default=None)
@click.option("--build/--no-build", "build_", default=True)
@pass_ctx
def team_init_cmd(ctx, file_, teams_root, build_):
    from kukeon_amd.teams.pipeline import team_init
    try:
        res = team_init(ctx._controller(), file_, realm=ctx.realm,
                        space=ctx.space, teams_root=teams_root,
                        build_images=build_)
    except errors.KukeonError as e:
        _die(e)
    click.echo(f"team {res['team']} (source {res['source']})")
    for kind, name, action in res["applied"]:
        click.echo(f"  {kind.lower()}/{name} {action}")
    for ref in res["built"]:
        click.echo(f"  image kukeon.internal/{ref} registered")
    for p in res["pruned"]:
        click.echo(f"  pruned {p}")


@cli.group()
def image():
    """Image catalog verbs (always in-process, like the reference)."""


@image.command("list")
@pass_ctx
def image_list(ctx):
    for img in ctx._controller().list_images():
        click.echo(img["name"])


@image.command("get")
@click.argument("name")
@pass_ctx
def image_get(ctx, name):
    try:
        click.echo(json.dumps(ctx._controller().get_image(name), indent=2))
    except errors.KukeonError as e:
        _die(e)


@image.command("delete")
@click.argument("name")
@click.option("--force", is_flag=True,
              help="delete even while a cell spec references the image")
@pass_ctx
def image_delete(ctx, name, force):
    try:
        ctx._controller().delete_image(name, force=force)
        click.echo(f"image {name} deleted")
    except errors.KukeonError as e:
        _die(e)


@image.command("prune")
@pass_ctx
def image_prune(ctx):
    for name in ctx._controller().prune_images():
        click.echo(f"pruned {name}")


@cli.command()
@click.argument("context", type=click.Path(exists=True), required=False)
@click.option("-t", "--tag", "tag", required=True, help="image name")
@click.option("-f", "--file", "file_", type=click.Path(exists=True),
              help="Kukefile (default <context>/Kukefile); a YAML file "
                   "falls back to profile registration")
@pass_ctx
def build(ctx, context, tag, file_):
    """Build a layered rootfs image from a Kukefile (kukebuild analog:
    FROM/COPY/RUN/ENV/CMD/WORKDIR/LABEL executed natively with
    overlayfs layers — reference cmd/kukebuild/main.go:17-49). With a
    YAML -f and no context, registers a runtime profile instead."""
    from kukeon_amd.images import Builder, ImageStore
    ctl = ctx._controller()
    if context is None:
        if not file_:
            raise click.UsageError("need a build context or -f profile")
        spec = yaml.safe_load(Path(file_).read_text()) or {}
        ctl.register_image(tag, spec)
        click.echo(f"image {tag} registered (profile)")
        return
    kf = Path(file_) if file_ else Path(context) / "Kukefile"
    if not kf.exists():
        raise click.UsageError(f"no Kukefile at {kf}")
    store = ImageStore(str(ctl.run_path))
    man = Builder(store).build(Path(context), kf.read_text(), tag,
                               log=lambda m: click.echo(m))
    click.echo(f"image {tag}: {len(man['layers'])} layer(s), "
               f"{man['sizeBytes']} bytes")


@cli.group()
def daemon():
    """Daemon management."""


@daemon.command("serve")
@click.option("--reconcile-interval", default=None, type=float)
@click.option("--configuration", default=None,
              type=click.Path(exists=True),
              help="ServerConfiguration YAML")
@click.option("--foreground/--no-foreground", default=True)
@pass_ctx
def daemon_serve(ctx, reconcile_interval, configuration, foreground):
    import logging

    from kukeon_amd.utils import logging as klog
    klog.setup(logging.INFO)
    server_cfg = None
    if configuration:
        raw = yaml.safe_load(Path(configuration).read_text())
        doc = api.ServerConfigurationDoc.from_dict(raw)
        server_cfg = doc.spec
    if reconcile_interval is None:
        reconcile_interval = (server_cfg.reconcile_interval_seconds
                              if server_cfg else 30.0)
    ctl = ctx._controller(server_cfg)
    ctl.bootstrap()
    from kukeon_amd.daemon.server import Server
    from kukeon_amd.runtime import sysuser
    srv = Server(ctl, ctx.socket_path, reconcile_interval,
                 socket_gid=sysuser.lookup_group())
    srv.start()
    if foreground:
        srv.wait()


@daemon.command("stop")
@pass_ctx
def daemon_stop(ctx):
    try:
        ctx.client.DaemonStop()
        click.echo("daemon stopping")
    except errors.KukeonError as e:
        _die(e)


@daemon.command("recreate")
@click.option("--model", default="llama-3-8b")
@click.option("--gpus", type=int, default=1)
@pass_ctx
def daemon_recreate(ctx, model, gpus):
    """Re-provision the system modelhub cell (reference `kuke daemon
    recreate` re-provisions the containerized kukeond cell via
    ProvisionKukeondCell, controller.go:253-280 — here the system cell
    is the inference server)."""
    from kukeon_amd.controller import naming
    ctl = ctx._controller()
    try:
        ctl.delete_cell(naming.SYSTEM_REALM, naming.SYSTEM_SPACE,
                        naming.SYSTEM_STACK, "modelhub", force=True)
    except errors.NotFound:
        pass
    doc = ctl.provision_modelhub_cell(model=model, gpus=gpus)
    click.echo(f"modelhub cell recreated: {doc.status.state}")


@daemon.command("status")
@pass_ctx
def daemon_status(ctx):
    try:
        st = ctx.client.Status()
        click.echo(json.dumps(st, indent=2))
    except Exception as e:
        _die(errors.KukeonError(f"daemon unreachable: {e}"))


@cli.command()
@click.option("--yes", is_flag=True, help="skip confirmation")
@pass_ctx
def uninstall(ctx, yes):
    """Full teardown: kill cells, remove the run path state and cgroups."""
    if not yes:
        click.confirm(f"remove all kukeon state under {ctx.run_path}?",
                      abort=True)
    ctl = ctx._controller()
    for realm in list(ctl.store.list_children(ctl.store.data_root)):
        try:
            ctl.delete_realm(realm, cascade=True)
        except Exception as e:
            click.echo(f"warn: realm {realm}: {e}", err=True)
    import shutil as sh
    for sub in ("data", "sessions", "images", "persist", "s", "bin"):
        sh.rmtree(Path(ctx.run_path) / sub, ignore_errors=True)
    for f in ("gpus.json", "kukeond.pid", ".kukeon-instance.json",
              "kukeond.sock", "modelhub.sock"):
        with __import__("contextlib").suppress(OSError):
            (Path(ctx.run_path) / f).unlink()
    ctl.cgroups.delete("")
    click.echo(f"kukeon state removed from {ctx.run_path}")


@cli.command()
@click.argument("shell", type=click.Choice(["bash", "zsh", "fish"]),
                default="bash")
def autocomplete(shell):
    """Print shell completion setup for kuke."""
    var = {"bash": "bash_source", "zsh": "zsh_source",
           "fish": "fish_source"}[shell]
    click.echo(f'eval "$(_KUKE_COMPLETE={var} kuke)"')


@cli.command()
@click.option("-f", "--file", "files", multiple=True, required=True,
              type=click.Path(exists=True, allow_dash=True))
@pass_ctx
def create(ctx, files):
    """Create resources from manifests (create-only: fails on existing)."""
    for fp in files:
        text = sys.stdin.read() if fp == "-" else Path(fp).read_text()
        from kukeon_amd.controller import parser as docparser
        for doc in docparser.parse_documents(text):
            method = {"Realm": "CreateRealm", "Space": "CreateSpace",
                      "Stack": "CreateStack", "Cell": "CreateCell",
                      "Session": "CreateSession"}.get(doc.kind)
            try:
                if method:
                    ctx.client.call(method, doc=doc.to_dict())
                else:
                    ctx.client.ApplyDocuments(
                        yaml=yaml.safe_dump(doc.to_dict()))
                click.echo(f"{doc.kind.lower()}/{doc.metadata.name} created")
            except errors.KukeonError as e:
                _die(e)


@cli.command()
@pass_ctx
def refresh(ctx):
    """Re-derive all statuses from live runtime state."""
    try:
        res = ctx.client.RefreshAll()
    except errors.KukeonError as e:
        _die(e)
    click.echo(f"refreshed: {res}")


@cli.command()
def version():
    import kukeon_amd
    click.echo(f"kuke {kukeon_amd.__version__} (MI355X-native)")


def main():
    prog = os.path.basename(sys.argv[0])
    if prog == "kukeond":
        sys.argv.insert(1, "daemon")
        if len(sys.argv) == 2:
            sys.argv.append("serve")
    cli(prog_name="kuke")


if __name__ == "__main__":
    main()
