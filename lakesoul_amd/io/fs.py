"""Filesystem / object-store layer with read-through disk cache.

MI355X-native analog of the reference's L0 storage layer
(``rust/lakesoul-io/src/object_store.rs`` + ``src/cache/``):

- local paths pass straight through (the engine mmaps them);
- remote schemes (``s3://``, ``mock://``) are localized through a disk
  LRU cache before the native reader touches them — the reference's
  ``ReadThroughCache<DiskCache>`` (cache/read_through.rs:23,
  disk_cache.rs:92), at whole-object granularity (the native reader then
  does its own ranged access via mmap);
- writes land locally then upload (multipart-upload analog,
  multipart_writer.rs:43); aborts delete the partial upload.

``s3://`` uses pyarrow.fs.S3FileSystem configured from the same
Hadoop-style options the reference accepts (fs.s3a.endpoint /
access.key / secret.key — object_store.rs:22-82) or their env
equivalents. ``mock://`` maps to a local directory (set
LAKESOUL_MOCK_FS_ROOT) and exists so the remote code path is testable in
network-less CI.

Cache size: LAKESOUL_CACHE_SIZE bytes (reference cache/mod.rs:17-52),
default 10 GiB, at LAKESOUL_CACHE_DIR (default ~/.lakesoul/cache).
"""

from __future__ import annotations

import hashlib
import json
import os
import shutil
import threading
import time
from typing import Optional, Tuple


def _split_scheme(path: str) -> Tuple[str, str]:
    if "://" in path:
        scheme, rest = path.split("://", 1)
        return scheme, rest
    return "file", path


def is_remote(path: str) -> bool:
    return _split_scheme(path)[0] not in ("file",)


class DiskCache:
    """Whole-object LRU cache on local disk."""

    def __init__(self, root: Optional[str] = None, capacity: Optional[int] = None):
        self.root = root or os.environ.get(
            "LAKESOUL_CACHE_DIR", os.path.expanduser("~/.lakesoul/cache")
        )
        self.capacity = capacity or int(
            os.environ.get("LAKESOUL_CACHE_SIZE", str(10 * 1024**3))
        )
        os.makedirs(self.root, exist_ok=True)
        self._lock = threading.Lock()

    def _key(self, path: str) -> str:
        return hashlib.sha256(path.encode()).hexdigest()[:32]

    def local_path(self, path: str) -> str:
        return os.path.join(self.root, self._key(path))

    def get(self, path: str) -> Optional[str]:
        lp = self.local_path(path)
        if os.path.exists(lp):
            os.utime(lp, None)  # LRU touch
            return lp
        return None

    def put_from(self, path: str, src_local: str) -> str:
        lp = self.local_path(path)
        shutil.move(src_local, lp)
        self._evict()
        return lp

    def _evict(self) -> None:
        with self._lock:
            entries = []
            total = 0
            for name in os.listdir(self.root):
                p = os.path.join(self.root, name)
                try:
                    st = os.stat(p)
                except OSError:
                    continue
                entries.append((st.st_atime, st.st_size, p))
                total += st.st_size
            if total <= self.capacity:
                return
            entries.sort()
            for _, size, p in entries:
                try:
                    os.remove(p)
                    total -= size
                except OSError:
                    pass
                if total <= self.capacity:
                    break

    def stats(self) -> dict:
        files = os.listdir(self.root)
        return {
            "entries": len(files),
            "bytes": sum(
                os.path.getsize(os.path.join(self.root, f)) for f in files
            ),
            "capacity": self.capacity,
        }


def retry_io(fn, attempts: int = 4, base_delay: float = 0.1,
             retryable=(OSError, IOError)):
    """Exponential-backoff retry for remote IO (reference
    object_store.rs:22-82 configures the same via fs.s3a.* options).
    Jittered delays base*2^k; the last failure propagates."""
    import random as _random
    import time as _time

    last = None
    for k in range(attempts):
        try:
            return fn()
        except retryable as e:
            last = e
            if k + 1 < attempts:
                _time.sleep(base_delay * (2 ** k) * (0.5 + _random.random()))
    raise last


class FileSystem:
    """Scheme-dispatching filesystem with read-through localization."""

    def __init__(self, cache: Optional[DiskCache] = None, options: Optional[dict] = None):
        self.cache = cache
        self.options = options or {}
        self._s3 = None

    def _retry_attempts(self) -> int:
        # Hadoop-style option name, as the reference accepts
        return int(self.options.get(
            "fs.s3a.retry.limit", os.environ.get("LAKESOUL_S3_RETRIES", "4")))

    def _get_cache(self) -> DiskCache:
        if self.cache is None:
            self.cache = DiskCache()
        return self.cache

    # -- remote backends ------------------------------------------------ #

    def _mock_root(self) -> str:
        root = os.environ.get("LAKESOUL_MOCK_FS_ROOT")
        if not root:
            raise RuntimeError("mock:// needs LAKESOUL_MOCK_FS_ROOT")
        return root

    def _s3fs(self):
        if self._s3 is None:
            from pyarrow import fs as pafs  # pyarrow ships S3 support

            # Hadoop-style option names, as the reference accepts
            # (object_store.rs:22-82)
            kwargs = {}
            opt = self.options
            if opt.get("fs.s3a.endpoint") or os.environ.get("AWS_ENDPOINT"):
                kwargs["endpoint_override"] = opt.get(
                    "fs.s3a.endpoint", os.environ.get("AWS_ENDPOINT")
                )
            if opt.get("fs.s3a.access.key"):
                kwargs["access_key"] = opt["fs.s3a.access.key"]
                kwargs["secret_key"] = opt.get("fs.s3a.secret.key", "")
            self._s3 = pafs.S3FileSystem(**kwargs)
        return self._s3

    # -- operations ----------------------------------------------------- #

    def localize(self, path: str) -> str:
        """Return a local path for reading (through the disk cache for
        remote objects)."""
        scheme, rest = _split_scheme(path)
        if scheme == "file":
            return rest
        cache = self._get_cache()
        hit = cache.get(path)
        if hit:
            return hit
        tmp = cache.local_path(path) + ".part"
        if scheme == "mock":
            shutil.copyfile(os.path.join(self._mock_root(), rest), tmp)
        elif scheme in ("s3", "s3a"):
            def _dl():
                with self._s3fs().open_input_stream(rest) as src, open(tmp, "wb") as dst:
                    shutil.copyfileobj(src, dst)

            retry_io(_dl, attempts=self._retry_attempts())
        else:
            raise ValueError(f"unsupported scheme {scheme}://")
        return cache.put_from(path, tmp)

    def upload(self, local_path: str, dest: str) -> None:
        scheme, rest = _split_scheme(dest)
        if scheme == "file":
            if local_path != rest:
                shutil.move(local_path, rest)
            return
        if scheme == "mock":
            target = os.path.join(self._mock_root(), rest)
            os.makedirs(os.path.dirname(target), exist_ok=True)
            shutil.copyfile(local_path, target)
            return
        if scheme in ("s3", "s3a"):
            def _ul():
                with open(local_path, "rb") as src, self._s3fs().open_output_stream(rest) as dst:
                    shutil.copyfileobj(src, dst)

            retry_io(_ul, attempts=self._retry_attempts())
            return
        raise ValueError(f"unsupported scheme {scheme}://")

    def open_multipart(self, dest: str) -> "MultipartSink":
        """Streaming upload sink: parts written incrementally become
        visible only on complete(); abort() leaves no visible object
        (reference multipart_writer.rs:43,239 abort semantics)."""
        return MultipartSink(self, dest)

    def delete(self, path: str) -> None:
        scheme, rest = _split_scheme(path)
        if scheme == "file":
            if os.path.exists(rest):
                os.remove(rest)
        elif scheme == "mock":
            p = os.path.join(self._mock_root(), rest)
            if os.path.exists(p):
                os.remove(p)
        elif scheme in ("s3", "s3a"):
            self._s3fs().delete_file(rest)

    def makedirs(self, path: str) -> None:
        scheme, rest = _split_scheme(path)
        if scheme == "file":
            os.makedirs(rest, exist_ok=True)
        elif scheme == "mock":
            os.makedirs(os.path.join(self._mock_root(), rest), exist_ok=True)
        # s3: directories are implicit


class MultipartSink:
    """Incremental writer to any scheme. The in-progress object is
    invisible until complete():

    - file:// and mock://: parts append to a hidden ``.__inprogress``
      sibling; complete() atomically renames it into place; abort()
      removes it (no final object ever exists).
    - s3://: parts stream into pyarrow's S3 output stream (multipart
      upload under the hood — S3 itself keeps parts invisible until the
      upload completes); abort() closes and best-effort-deletes.
    """

    def __init__(self, fs: FileSystem, dest: str):
        self.fs = fs
        self.dest = dest
        self.parts = 0
        self.bytes = 0
        self._done = False
        scheme, rest = _split_scheme(dest)
        self._scheme = scheme
        if scheme == "file":
            self._tmp = rest + ".__inprogress"
            os.makedirs(os.path.dirname(rest), exist_ok=True)
            self._f = open(self._tmp, "wb")
        elif scheme == "mock":
            target = os.path.join(fs._mock_root(), rest)
            os.makedirs(os.path.dirname(target), exist_ok=True)
            self._final = target
            self._tmp = target + ".__inprogress"
            self._f = open(self._tmp, "wb")
        elif scheme in ("s3", "s3a"):
            self._f = fs._s3fs().open_output_stream(rest)
            self._tmp = None
        else:
            raise ValueError(f"unsupported scheme {scheme}://")

    def write_part(self, data: bytes) -> None:
        assert not self._done
        self._f.write(data)
        self.parts += 1
        self.bytes += len(data)

    def complete(self) -> None:
        if self._done:
            return
        self._done = True
        self._f.close()
        if self._scheme == "file":
            os.replace(self._tmp, _split_scheme(self.dest)[1])
        elif self._scheme == "mock":
            os.replace(self._tmp, self._final)

    def abort(self) -> None:
        """Cancel: no visible object remains."""
        if self._done:
            return
        self._done = True
        try:
            self._f.close()
        except Exception:
            pass
        if self._tmp is not None and os.path.exists(self._tmp):
            os.remove(self._tmp)
        if self._scheme in ("s3", "s3a"):
            try:
                self.fs.delete(self.dest)
            except Exception:
                pass


_default_fs: Optional[FileSystem] = None


def default_fs() -> FileSystem:
    global _default_fs
    if _default_fs is None:
        _default_fs = FileSystem()
    return _default_fs
