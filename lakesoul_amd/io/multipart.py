"""Streaming multipart parquet upload: row-group encode overlapped with
part upload (reference multipart_writer.rs:43 — encode the next row
group while the previous part ships), with abort that cancels the upload
so no partial object ever becomes visible (async_writer/mod.rs:80-85).

The C++ incremental writer (module.cc writer_open/write/finish) appends
row groups to a local spool with a stable byte boundary after every
write; an uploader thread streams [uploaded, boundary) to the
MultipartSink while the next chunk encodes.
"""

from __future__ import annotations

import os
import queue
import tempfile
import threading
from typing import List, Optional

import torch

from ..ops import cpp
from .batch import Batch
from .fs import default_fs

_CODEC_ID = {"zstd": 6, "none": 0, "uncompressed": 0}


class StreamingParquetUpload:
    """Encode Batch chunks into parquet row groups and upload finished
    byte ranges concurrently."""

    def __init__(self, dest: str, schema, compression: str = "zstd",
                 level: int = 1, row_group_size: int = 250_000,
                 part_bytes: int = 8 << 20):
        self.dest = dest
        self.schema = schema
        self.part_bytes = part_bytes
        fd, self._spool = tempfile.mkstemp(suffix=".parquet.spool")
        os.close(fd)
        from .writer import expand_schema_leaves

        names, dtypes, nullable, parents = expand_schema_leaves(schema)
        self._h = cpp().writer_open(self._spool, names, dtypes, nullable,
                                    row_group_size,
                                    _CODEC_ID.get(compression, 6), level,
                                    parents)
        self._sink = default_fs().open_multipart(dest)
        self._uploaded = 0
        self._rows = 0
        self._err: Optional[BaseException] = None
        self._q: "queue.Queue" = queue.Queue(maxsize=4)
        self._up = threading.Thread(target=self._upload_loop, daemon=True)
        self._up.start()
        self._closed = False
        self.upload_events: List[int] = []  # byte boundary per shipped part

    # -- uploader thread ------------------------------------------------ #

    def _upload_loop(self):
        try:
            with open(self._spool, "rb") as f:
                while True:
                    upto = self._q.get()
                    if upto is None:
                        return
                    while self._uploaded < upto:
                        n = min(self.part_bytes, upto - self._uploaded)
                        f.seek(self._uploaded)
                        data = f.read(n)
                        if not data:
                            break
                        self._sink.write_part(data)
                        self._uploaded += len(data)
                        self.upload_events.append(self._uploaded)
        except BaseException as e:  # surfaced on the writer thread
            self._err = e

    def _check(self):
        if self._err is not None:
            raise RuntimeError(f"upload failed: {self._err}") from self._err

    # -- writer side ----------------------------------------------------- #

    def write_batch(self, batch: Batch) -> None:
        """Encode one chunk as row group(s); finished bytes are handed to
        the uploader (overlaps with the caller's next chunk prep)."""
        self._check()
        from .writer import marshal_batch

        (_, _, cols, offs, vals, _, eoffs, _) = marshal_batch(batch)
        cpp().writer_write(self._h, cols, offs, vals, eoffs)
        self._rows += batch.num_rows
        self._q.put(cpp().writer_bytes(self._h))

    def close(self) -> int:
        """Finish the footer, ship the tail, finalize the upload.
        Returns total file size."""
        self._check()
        size = cpp().writer_finish(self._h)
        self._q.put(size)
        self._q.put(None)
        self._up.join(timeout=300)
        self._check()
        self._sink.complete()
        self._closed = True
        os.remove(self._spool)
        return size

    def abort(self) -> None:
        """Cancel everything: upload aborted (no visible object), spool
        removed."""
        if self._closed:
            return
        self._closed = True
        try:
            self._q.put(None)
            self._up.join(timeout=60)
        except Exception:
            pass
        try:
            cpp().writer_abort(self._h)
        except Exception:
            pass
        self._sink.abort()
        if os.path.exists(self._spool):
            os.remove(self._spool)

    @property
    def rows(self) -> int:
        return self._rows
