"""Distributed key-value store for rendezvous.

Reference semantics: realhf/base/name_resolve.py (NameRecordRepository:32,
MemoryNameRecordRepository:181, NfsNameRecordRepository:265).  Used to
exchange addresses/metadata between processes before torch.distributed is
up (e.g. master addr/port, per-worker GPU claims).

The file-backed store works over any shared filesystem (single node:
/tmp; cluster: NFS).  A Redis backend (reference:
RedisNameRecordRepository, name_resolve.py:84) is provided for clusters
without a shared FS — select with REALHF_AMD_NAME_RESOLVE=redis and
REALHF_AMD_REDIS_ADDR=host:port (requires the redis-py client).
"""
import os
import shutil
import time
from typing import List, Optional


class NameEntryExistsError(Exception):
    pass


class NameEntryNotFoundError(Exception):
    pass


class MemoryNameRecordRepository:
    def __init__(self):
        self._store = {}

    def add(self, name, value, replace=False, **kw):
        if not replace and name in self._store:
            raise NameEntryExistsError(name)
        self._store[name] = str(value)

    def get(self, name) -> str:
        try:
            return self._store[name]
        except KeyError:
            raise NameEntryNotFoundError(name)

    def delete(self, name):
        self._store.pop(name, None)

    def get_subtree(self, prefix) -> List[str]:
        return [v for k, v in sorted(self._store.items()) if k.startswith(prefix)]

    def find_subtree(self, prefix) -> List[str]:
        return sorted(k for k in self._store if k.startswith(prefix))

    def clear_subtree(self, prefix):
        for k in list(self._store):
            if k.startswith(prefix):
                del self._store[k]

    def wait(self, name, timeout=60, poll=0.05) -> str:
        t0 = time.monotonic()
        while True:
            try:
                return self.get(name)
            except NameEntryNotFoundError:
                if time.monotonic() - t0 > timeout:
                    raise TimeoutError(f"name_resolve.wait timed out on {name}")
                time.sleep(poll)


class FileNameRecordRepository(MemoryNameRecordRepository):
    """Names are files under `root`; '/' in the name maps to directories.
    Atomic add via O_EXCL; values are small strings."""

    def __init__(self, root: Optional[str] = None):
        self.root = root or os.environ.get(
            "REALHF_AMD_NAME_RESOLVE_ROOT", "/tmp/realhf_amd_name_resolve"
        )
        os.makedirs(self.root, exist_ok=True)

    def _path(self, name):
        return os.path.join(self.root, name.strip("/"))

    def add(self, name, value, replace=False, **kw):
        p = self._path(name)
        os.makedirs(os.path.dirname(p), exist_ok=True)
        flags = os.O_WRONLY | os.O_CREAT | (0 if replace else os.O_EXCL)
        try:
            fd = os.open(p + ".tmp" if replace else p, flags, 0o644)
        except FileExistsError:
            raise NameEntryExistsError(name)
        with os.fdopen(fd, "w") as f:
            f.write(str(value))
        if replace:
            os.replace(p + ".tmp", p)

    def get(self, name) -> str:
        try:
            with open(self._path(name)) as f:
                return f.read()
        except FileNotFoundError:
            raise NameEntryNotFoundError(name)

    def delete(self, name):
        try:
            os.remove(self._path(name))
        except FileNotFoundError:
            pass

    def find_subtree(self, prefix) -> List[str]:
        base = self._path(prefix)
        out = []
        if os.path.isdir(base):
            for dirpath, _, files in os.walk(base):
                for fn in files:
                    rel = os.path.relpath(os.path.join(dirpath, fn), self.root)
                    out.append(rel)
        elif os.path.isfile(base):
            out.append(prefix.strip("/"))
        return sorted(out)

    def get_subtree(self, prefix) -> List[str]:
        return [self.get(k) for k in self.find_subtree(prefix)]

    def clear_subtree(self, prefix):
        base = self._path(prefix)
        if os.path.isdir(base):
            shutil.rmtree(base, ignore_errors=True)
        elif os.path.isfile(base):
            os.remove(base)


class RedisNameRecordRepository(MemoryNameRecordRepository):
    """Redis-backed store (reference: name_resolve.py:84).  Keys carry an
    optional TTL via keepalive; values are strings.  The client object is
    injectable for testing (any object with set/get/delete/scan_iter)."""

    def __init__(self, addr: Optional[str] = None, client=None,
                 ttl_secs: Optional[int] = None):
        if client is None:
            import redis  # requires redis-py (not bundled in this image)

            host, _, port = (addr or os.environ.get(
                "REALHF_AMD_REDIS_ADDR", "localhost:6379")).partition(":")
            client = redis.Redis(host=host, port=int(port or 6379))
        self._r = client
        self._ttl = ttl_secs

    def add(self, name, value, replace=False, **kw):
        if not replace and self._r.get(name) is not None:
            raise NameEntryExistsError(name)
        self._r.set(name, str(value), ex=self._ttl)

    def get(self, name) -> str:
        v = self._r.get(name)
        if v is None:
            raise NameEntryNotFoundError(name)
        return v.decode() if isinstance(v, bytes) else str(v)

    def delete(self, name):
        self._r.delete(name)

    def find_subtree(self, prefix) -> List[str]:
        keys = [k.decode() if isinstance(k, bytes) else str(k)
                for k in self._r.scan_iter(match=prefix + "*")]
        return sorted(keys)

    def get_subtree(self, prefix) -> List[str]:
        return [self.get(k) for k in self.find_subtree(prefix)]

    def clear_subtree(self, prefix):
        for k in self.find_subtree(prefix):
            self._r.delete(k)


_default = None


def default_repository() -> MemoryNameRecordRepository:
    global _default
    if _default is None:
        if os.environ.get("REALHF_AMD_NAME_RESOLVE") == "redis":
            _default = RedisNameRecordRepository()
        else:
            _default = FileNameRecordRepository()
    return _default


def add(name, value, replace=False):
    default_repository().add(name, value, replace=replace)


def get(name):
    return default_repository().get(name)


def wait(name, timeout=60):
    return default_repository().wait(name, timeout=timeout)


def clear_subtree(prefix):
    default_repository().clear_subtree(prefix)


def get_subtree(prefix):
    return default_repository().get_subtree(prefix)


# -- key schema (reference: realhf/base/names.py) ---------------------------
def trial_root(experiment, trial):
    return f"trials/{experiment}/{trial}"


def distributed_peer(experiment, trial, peer_idx):
    return f"{trial_root(experiment, trial)}/peers/{peer_idx}"


def master_addr(experiment, trial):
    return f"{trial_root(experiment, trial)}/master_addr"
