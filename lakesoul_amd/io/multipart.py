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
        names = [f.name for f in schema]
        dtypes = [f.dtype for f in schema]
        nullable = [f.nullable for f in schema]
        self._h = cpp().writer_open(self._spool, names, dtypes, nullable,
                                    row_group_size,
                                    _CODEC_ID.get(compression, 6), level)
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
        cols, offs, vals = [], [], []
        eoffs = []
        for f in self.schema:
            c = batch.columns[f.name]
            eoffs.append(None)
            if c.is_list_str:
                cols.append(c.bytes_.cpu())
                offs.append(c.offsets.cpu().to(torch.int64))
                eoffs[-1] = c.elem_offsets.cpu().to(torch.int32)
            elif c.is_list:
                t = c.data.cpu()
                if f.dtype[5:-1] in ("int8", "int16"):
                    t = t.to(torch.int32)
                cols.append(t)
                offs.append(c.offsets.cpu().to(torch.int64))
            elif c.is_string:
                cols.append(c.bytes_.cpu())
                offs.append(c.offsets.cpu())
            else:
                t = c.data.cpu()
                if f.dtype in ("int8", "int16"):
                    t = t.to(torch.int32)
                if f.dtype == "bool":
                    t = t.to(torch.uint8)
                cols.append(t)
                offs.append(None)
            vals.append(None if c.validity is None else c.validity.cpu())
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
