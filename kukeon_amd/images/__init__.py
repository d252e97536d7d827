"""Layered image store + builder (the kukebuild analog).

Reference: cmd/kukebuild embeds BuildKit and writes OCI images into the
realm's containerd namespace (cmd/kukebuild/main.go:17-49). This
environment has no container engine, so images are layered rootfs trees
built natively:

* a LAYER is an extracted directory `<run>/layers/<id>/root/` plus its
  tar `<run>/layers/<id>.tar` (content-addressed by the tar's sha256),
* an IMAGE is a manifest `<run>/images/<safe-name>.json` naming an
  ordered layer list and a config (env/cmd/workdir),
* `Builder` executes a Kukefile (FROM / COPY / RUN / ENV / CMD /
  WORKDIR / LABEL): each COPY or RUN step materializes the current
  rootfs as an overlayfs mount (lower = base layers) inside a private
  mount namespace, applies the step (RUN = chroot'ed `sh -c`), and
  commits the overlay upperdir as a new layer,
* at cell start the runner hands the shim the layer paths; the shim
  overlay-mounts them (upper = per-container scratch) in ITS mount
  namespace and chroots — the mount dies with the container.

Hosts without overlayfs degrade: single-layer images chroot straight
into a copied tree; builds refuse RUN steps.
"""
from __future__ import annotations

import hashlib
import json
import os
import shutil
import subprocess
import tarfile
import tempfile
import time
from pathlib import Path
from typing import Dict, List, Optional

from kukeon_amd.api import errors

LAYER_DIR = "layers"
IMAGE_DIR = "images"


def _now() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())


class ImageStore:
    def __init__(self, run_path: str):
        self.run = Path(run_path)
        (self.run / LAYER_DIR).mkdir(parents=True, exist_ok=True)
        (self.run / IMAGE_DIR).mkdir(parents=True, exist_ok=True)

    # -- layers --------------------------------------------------------
    def add_layer_from_dir(self, src: Path) -> str:
        """Pack `src` into a content-addressed layer (tar + extracted
        tree); returns the layer id."""
        with tempfile.NamedTemporaryFile(dir=self.run / LAYER_DIR,
                                         suffix=".tar",
                                         delete=False) as tmp:
            with tarfile.open(tmp.name, "w") as tf:
                for entry in sorted(src.rglob("*")):
                    tf.add(entry, arcname=str(entry.relative_to(src)),
                           recursive=False)
            tmp.flush()
            digest = hashlib.sha256(Path(tmp.name).read_bytes()).hexdigest()
            lid = digest[:24]
            final_tar = self.run / LAYER_DIR / f"{lid}.tar"
            froot = self.run / LAYER_DIR / lid / "root"
            if final_tar.exists():
                os.unlink(tmp.name)
                return lid
            os.replace(tmp.name, final_tar)
        froot.parent.mkdir(parents=True, exist_ok=True)
        shutil.copytree(src, froot, symlinks=True, dirs_exist_ok=True)
        return lid

    def layer_root(self, lid: str) -> Path:
        p = self.run / LAYER_DIR / lid / "root"
        if not p.is_dir():
            raise errors.NotFound(f"layer {lid}")
        return p

    # -- images --------------------------------------------------------
    def _manifest_path(self, name: str) -> Path:
        return self.run / IMAGE_DIR / (name.replace("/", "_") + ".json")

    def put_manifest(self, name: str, layers: List[str],
                     config: Optional[Dict] = None,
                     labels: Optional[Dict] = None) -> Dict:
        man = {
            "name": name,
            "layers": layers,
            "config": config or {},
            "labels": labels or {},
            "createdAt": _now(),
            "sizeBytes": sum(
                (self.run / LAYER_DIR / f"{l}.tar").stat().st_size
                for l in layers),
        }
        p = self._manifest_path(name)
        p.parent.mkdir(parents=True, exist_ok=True)
        tmp = p.with_suffix(".tmp")
        tmp.write_text(json.dumps(man, indent=2))
        os.replace(tmp, p)
        return man

    def get(self, name: str) -> Dict:
        p = self._manifest_path(name)
        if not p.exists():
            raise errors.NotFound(f"image {name}")
        return json.loads(p.read_text())

    def exists(self, name: str) -> bool:
        return self._manifest_path(name).exists()

    def layer_paths(self, name: str) -> List[Path]:
        return [self.layer_root(l) for l in self.get(name)["layers"]]

    def delete(self, name: str) -> None:
        p = self._manifest_path(name)
        if not p.exists():
            raise errors.NotFound(f"image {name}")
        p.unlink()

    def prune_layers(self) -> List[str]:
        """Drop layers referenced by no manifest."""
        used = set()
        for mp in (self.run / IMAGE_DIR).glob("*.json"):
            try:
                used.update(json.loads(mp.read_text()).get("layers", []))
            except (ValueError, OSError):
                continue
        dropped = []
        for tar in (self.run / LAYER_DIR).glob("*.tar"):
            lid = tar.stem
            if lid not in used:
                tar.unlink()
                shutil.rmtree(self.run / LAYER_DIR / lid,
                              ignore_errors=True)
                dropped.append(lid)
        return dropped


def overlay_supported() -> bool:
    from kukeon_amd.runtime import namespaces as nsmod
    if not nsmod.can_unshare(nsmod.CLONE_NEWNS):
        return False
    probe = Path(tempfile.mkdtemp(prefix="kuke-ovl-"))
    try:
        for d in ("low", "up", "work", "mnt"):
            (probe / d).mkdir()
        pid = os.fork()
        if pid == 0:
            try:
                nsmod.unshare(nsmod.CLONE_NEWNS)
                nsmod.make_mounts_private()
                nsmod.mount(
                    "overlay", str(probe / "mnt"), "overlay", 0,
                    f"lowerdir={probe/'low'},upperdir={probe/'up'},"
                    f"workdir={probe/'work'}")
                os._exit(0)
            except OSError:
                os._exit(1)
        _, status = os.waitpid(pid, 0)
        return os.waitstatus_to_exitcode(status) == 0
    finally:
        shutil.rmtree(probe, ignore_errors=True)


class BuildError(errors.KukeonError):
    code = "ErrImageBuild"


class Builder:
    """Kukefile executor. Supported instructions:
    FROM <image|scratch>, COPY <src...> <dst>, RUN <cmd>, ENV K=V,
    CMD <cmd>, WORKDIR <dir>, LABEL K=V."""

    def __init__(self, store: ImageStore):
        self.store = store

    def build(self, context: Path, kukefile: str, tag: str,
              log=lambda s: None) -> Dict:
        from kukeon_amd.runtime import namespaces as nsmod
        lines = self._parse(kukefile)
        layers: List[str] = []
        config: Dict = {"env": [], "cmd": "", "workdir": ""}
        labels: Dict = {}
        if not lines or lines[0][0] != "FROM":
            raise BuildError("Kukefile must start with FROM")
        base = lines[0][1]
        if base != "scratch":
            man = self.store.get(base)  # NotFound raises through
            layers = list(man["layers"])
            config.update({k: v for k, v in man.get("config", {}).items()
                           if v})
        have_overlay = overlay_supported()
        for op, arg in lines[1:]:
            if op == "ENV":
                config["env"].append(arg)
            elif op == "CMD":
                config["cmd"] = arg
            elif op == "WORKDIR":
                config["workdir"] = arg
            elif op == "LABEL":
                k, _, v = arg.partition("=")
                labels[k] = v
            elif op == "COPY":
                parts = arg.split()
                srcs, dst = parts[:-1], parts[-1]
                stage = Path(tempfile.mkdtemp(prefix="kuke-copy-"))
                try:
                    ddir = stage / dst.lstrip("/")
                    for srel in srcs:
                        sp = (context / srel).resolve()
                        if not str(sp).startswith(str(context.resolve())):
                            raise BuildError(f"COPY source {srel} escapes "
                                             "the build context")
                        if sp.is_dir():
                            shutil.copytree(sp, ddir, symlinks=True,
                                            dirs_exist_ok=True)
                        else:
                            ddir.mkdir(parents=True, exist_ok=True)
                            shutil.copy2(sp, ddir / sp.name)
                    lid = self.store.add_layer_from_dir(stage)
                    layers.append(lid)
                    log(f"COPY -> layer {lid}")
                finally:
                    shutil.rmtree(stage, ignore_errors=True)
            elif op == "RUN":
                if not have_overlay:
                    raise BuildError(
                        "RUN requires overlayfs (unavailable on this "
                        "host); COPY-only builds still work")
                lid = self._run_step(arg, layers, config, log)
                layers.append(lid)
            else:
                raise BuildError(f"unsupported instruction {op}")
        man = self.store.put_manifest(tag, layers, config, labels)
        log(f"built {tag}: {len(layers)} layer(s), "
            f"{man['sizeBytes']} bytes")
        return man

    def _run_step(self, cmd: str, layers: List[str], config: Dict,
                  log) -> str:
        """Execute `sh -c cmd` chroot'ed into an overlay of the current
        layers; the upperdir becomes the new layer."""
        from kukeon_amd.runtime import namespaces as nsmod
        work = Path(tempfile.mkdtemp(prefix="kuke-run-"))
        (work / "up").mkdir()
        (work / "ovl").mkdir()
        (work / "w").mkdir()
        (work / "lo0").mkdir()  # FROM scratch + RUN-first: overlayfs
        # refuses upperdir as its own lowerdir, so give it an empty lower
        lowers = ":".join(str(self.store.layer_root(l))
                          for l in reversed(layers)) or str(work / "lo0")
        rc_r, rc_w = os.pipe()
        pid = os.fork()
        if pid == 0:
            os.close(rc_r)
            try:
                nsmod.unshare(nsmod.CLONE_NEWNS)
                nsmod.make_mounts_private()
                nsmod.mount("overlay", str(work / "ovl"), "overlay", 0,
                            f"lowerdir={lowers},upperdir={work/'up'},"
                            f"workdir={work/'w'}")
                os.chroot(str(work / "ovl"))
                os.chdir(config.get("workdir") or "/")
                env = dict(os.environ)
                for kv in config.get("env", []):
                    k, _, v = kv.partition("=")
                    env[k] = v
                p = subprocess.run(["/bin/sh", "-c", cmd], env=env,
                                   capture_output=True, timeout=600)
                os.write(rc_w, p.stderr[-500:] if p.returncode else b"")
                os._exit(p.returncode & 0xFF)
            except BaseException as e:  # noqa: BLE001
                try:
                    os.write(rc_w, str(e).encode()[:500])
                except OSError:
                    pass
                os._exit(126)
        os.close(rc_w)
        err = b""
        while True:
            chunk = os.read(rc_r, 4096)
            if not chunk:
                break
            err += chunk
        os.close(rc_r)
        _, status = os.waitpid(pid, 0)
        rc = os.waitstatus_to_exitcode(status)
        if rc != 0:
            shutil.rmtree(work, ignore_errors=True)
            raise BuildError(f"RUN {cmd!r} failed (rc={rc}): "
                             f"{err.decode('utf-8', 'replace')}")
        try:
            lid = self.store.add_layer_from_dir(work / "up")
            log(f"RUN {cmd!r} -> layer {lid}")
            return lid
        finally:
            shutil.rmtree(work, ignore_errors=True)

    @staticmethod
    def _parse(text: str):
        out = []
        cont = ""
        for raw in text.splitlines():
            line = raw.strip()
            if not line or line.startswith("#"):
                continue
            if cont:
                line = cont + " " + line
                cont = ""
            if line.endswith("\\"):
                cont = line[:-1].strip()
                continue
            op, _, arg = line.partition(" ")
            out.append((op.upper(), arg.strip()))
        if cont:
            out.append(tuple((cont.partition(" ")[0].upper(),
                              cont.partition(" ")[2].strip())))
        return out
