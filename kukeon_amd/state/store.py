"""On-disk metadata store: the durable state tree.

Layout (byte-layout contract from the reference — SURVEY.md §5 checkpoint
row; studied from /root/reference/internal/util/fs/metadata.go:39-527 and
internal/metadata/lock.go:41-193 — sidecar flock, CAS writes, K8s-style
Generation/ObservedGeneration from v1beta1/cell.go:39-42,216-219; socket
symlink dir for the SUN_PATH 107-byte limit from consts.go:105-124):

    <run_path>/data/<realm>/metadata.json
    <run_path>/data/<realm>/<space>/metadata.json
    <run_path>/data/<realm>/<space>/network.json
    <run_path>/data/<realm>/<space>/<stack>/metadata.json
    <run_path>/data/<realm>/<space>/<stack>/<cell>/metadata.json
    <run_path>/data/<realm>/<space>/<stack>/<cell>/<container>/metadata.json
    <run_path>/data/<realm>/<space>/secrets/<name>.json        (0600)
    <run_path>/data/<realm>/<space>/blueprints/<name>.json
    <run_path>/data/<realm>/<space>/configs/<name>.json
    <run_path>/data/<realm>/<space>/volumes/<name>/            (data dir)
    <run_path>/data/<realm>/<space>/volume-meta/<name>.json
    <run_path>/sessions/<realm>/<space>/<stack>/<name>.json
    <run_path>/s/<short>  -> socket symlinks (SUN_PATH 107-byte limit)

Concurrency: a sidecar `.lock` flock per document + compare-and-swap on
metadata.generation (StaleResource on a lost update), exactly the
reference's correctness envelope (metadata/lock.go WriteMetadataCAS).
"""
from __future__ import annotations

import contextlib
import fcntl
import hashlib
import json
import os
import tempfile
from pathlib import Path
from typing import Any, Dict, Iterator, List, Optional

from kukeon_amd.api import errors

METADATA_FILE = "metadata.json"
DATA_DIR = "data"
SESSIONS_DIR = "sessions"
SOCKET_LINK_DIR = "s"


class Store:
    def __init__(self, run_path: str):
        self.run_path = Path(run_path)

    # ---- path helpers ------------------------------------------------
    @property
    def data_root(self) -> Path:
        return self.run_path / DATA_DIR

    def realm_dir(self, realm: str) -> Path:
        return self.data_root / realm

    def space_dir(self, realm: str, space: str) -> Path:
        return self.realm_dir(realm) / space

    def stack_dir(self, realm: str, space: str, stack: str) -> Path:
        return self.space_dir(realm, space) / stack

    def cell_dir(self, realm: str, space: str, stack: str, cell: str) -> Path:
        return self.stack_dir(realm, space, stack) / cell

    def container_dir(self, realm, space, stack, cell, container) -> Path:
        return self.cell_dir(realm, space, stack, cell) / container

    def session_path(self, realm, space, stack, name) -> Path:
        return (self.run_path / SESSIONS_DIR / realm / space / stack /
                f"{name}.json")

    def scoped_doc_path(self, realm, space, kind_dir, name) -> Path:
        return self.space_dir(realm, space) / kind_dir / f"{name}.json"

    def volume_data_dir(self, realm, space, name) -> Path:
        return self.space_dir(realm, space) / "volumes" / name

    def socket_link(self, target: str) -> Path:
        """Short symlink for unix sockets whose real path exceeds SUN_PATH."""
        short = hashlib.sha256(target.encode()).hexdigest()[:12]
        d = self.run_path / SOCKET_LINK_DIR
        d.mkdir(parents=True, exist_ok=True)
        link = d / short
        with contextlib.suppress(FileNotFoundError):
            link.unlink()
        link.symlink_to(target)
        return link

    # ---- locking -----------------------------------------------------
    @contextlib.contextmanager
    def lock(self, path: Path) -> Iterator[None]:
        """Exclusive flock on the sidecar `<file>.lock`."""
        lock_path = Path(str(path) + ".lock")
        lock_path.parent.mkdir(parents=True, exist_ok=True)
        fd = os.open(lock_path, os.O_CREAT | os.O_RDWR, 0o644)
        try:
            fcntl.flock(fd, fcntl.LOCK_EX)
            yield
        finally:
            fcntl.flock(fd, fcntl.LOCK_UN)
            os.close(fd)

    # ---- document IO -------------------------------------------------
    def read(self, path: Path) -> Optional[Dict[str, Any]]:
        try:
            with open(path) as f:
                return json.load(f)
        except FileNotFoundError:
            return None

    def _atomic_write(self, path: Path, data: Dict[str, Any],
                      mode: int = 0o644) -> None:
        path.parent.mkdir(parents=True, exist_ok=True)
        fd, tmp = tempfile.mkstemp(dir=path.parent, prefix=".tmp-")
        try:
            with os.fdopen(fd, "w") as f:
                json.dump(data, f, indent=2, sort_keys=False)
                f.write("\n")
                f.flush()
                os.fsync(f.fileno())
            os.chmod(tmp, mode)
            os.replace(tmp, path)
        except BaseException:
            with contextlib.suppress(OSError):
                os.unlink(tmp)
            raise

    def write(self, path: Path, data: Dict[str, Any], mode: int = 0o644) -> None:
        """Plain locked write (no generation check)."""
        with self.lock(path):
            self._atomic_write(path, data, mode)

    def write_cas(self, path: Path, data: Dict[str, Any],
                  expected_generation: Optional[int] = None,
                  bump: bool = True, mode: int = 0o644) -> int:
        """Compare-and-swap write keyed on metadata.generation.

        Reads the current doc under the flock; if expected_generation is
        given and the on-disk generation differs, raises StaleResource (a
        lost optimistic update). On success writes with generation+1 (when
        bump) and returns the new generation.
        """
        with self.lock(path):
            cur = self.read(path)
            cur_gen = (cur or {}).get("metadata", {}).get("generation", 0)
            if expected_generation is not None and cur is not None \
                    and cur_gen != expected_generation:
                raise errors.StaleResource(
                    f"{path}: generation {cur_gen} != expected "
                    f"{expected_generation}")
            md = data.setdefault("metadata", {})
            new_gen = cur_gen + 1 if bump else cur_gen
            md["generation"] = new_gen
            self._atomic_write(path, data, mode)
            return new_gen

    def create_exclusive(self, path: Path, data: Dict[str, Any],
                         mode: int = 0o644) -> None:
        """Create-only write (reference: atomic os.Link create variant)."""
        with self.lock(path):
            if path.exists():
                raise errors.AlreadyExists(str(path))
            md = data.setdefault("metadata", {})
            md.setdefault("generation", 1)
            self._atomic_write(path, data, mode)

    def delete(self, path: Path) -> bool:
        # The sidecar .lock is left behind as a tombstone on purpose:
        # unlinking it while held would let a process already blocked in
        # flock() on the old inode acquire a stale lock concurrently with
        # a new holder that recreated the file — two writers in the CAS
        # critical section. delete_tree() reclaims them with
        # the directory.
        with self.lock(path):
            existed = path.exists()
            with contextlib.suppress(FileNotFoundError):
                path.unlink()
            return existed

    def delete_tree(self, d: Path) -> None:
        import shutil
        with contextlib.suppress(FileNotFoundError):
            shutil.rmtree(d)

    # ---- listing -----------------------------------------------------
    def list_children(self, d: Path) -> List[str]:
        if not d.is_dir():
            return []
        out = []
        for p in sorted(d.iterdir()):
            if p.is_dir() and (p / METADATA_FILE).exists():
                out.append(p.name)
        return out

    def list_scoped_docs(self, d: Path) -> List[str]:
        if not d.is_dir():
            return []
        return sorted(p.stem for p in d.glob("*.json"))
