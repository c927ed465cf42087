"""Chunked upload manager with a JSON-persisted registry.

Capability-parity with the reference's upload subsystem
(/root/reference/distllm/compute_node/uploads.py: sequential chunked
uploads, sha256 validation, on-disk layout, registry persisted on finalize
and restored at boot — SURVEY §5.4). Differences by design:

* thread-safe (one lock; the reference shares unsynchronized module
  singletons across server threads — SURVEY §5.2),
* sha256 streamed during upload instead of re-reading the file at the end,
* files are named by their metadata name (falling back to ``upload_<id>``),
  no funky-name generator.
"""
from __future__ import annotations

import hashlib
import json
import os
import threading
from dataclasses import dataclass
from typing import Dict, List, Optional


class UploadError(Exception):
    pass


class ParallelUploadError(UploadError):
    pass


class UploadNotFoundError(UploadError):
    pass


class FailedUploadError(UploadError):
    pass


@dataclass
class UploadRecord:
    upload_id: int
    kind: str                 # "slice" | "file"
    metadata: dict
    path: str
    size: int = 0
    checksum: str = ""
    finished: bool = False
    failed: bool = False

    def to_dict(self) -> dict:
        return dict(upload_id=self.upload_id, kind=self.kind,
                    metadata=self.metadata, path=self.path, size=self.size,
                    checksum=self.checksum, finished=self.finished,
                    failed=self.failed)

    @classmethod
    def from_dict(cls, d: dict) -> "UploadRecord":
        return cls(**d)


class UploadManager:
    """Sequential chunked uploads under ``root/{slices,files}/``."""

    REGISTRY_FILE = "registry.json"

    def __init__(self, root: str):
        self.root = root
        self._lock = threading.Lock()
        self._records: Dict[int, UploadRecord] = {}
        self._active: Optional[int] = None
        self._fh = None
        self._hasher = None
        self._next_id = 0
        os.makedirs(os.path.join(root, "slices"), exist_ok=True)
        os.makedirs(os.path.join(root, "files"), exist_ok=True)
        self._restore()

    # ------------------------------------------------------------ persist

    def _registry_path(self) -> str:
        return os.path.join(self.root, self.REGISTRY_FILE)

    def _persist(self) -> None:
        state = {
            "next_id": self._next_id,
            "records": [r.to_dict() for r in self._records.values()],
        }
        tmp = self._registry_path() + ".tmp"
        with open(tmp, "w") as f:
            json.dump(state, f, indent=1)
        os.replace(tmp, self._registry_path())

    def _restore(self) -> None:
        try:
            with open(self._registry_path()) as f:
                state = json.load(f)
        except FileNotFoundError:
            return
        self._next_id = state.get("next_id", 0)
        for d in state.get("records", []):
            r = UploadRecord.from_dict(d)
            self._records[r.upload_id] = r

    # ------------------------------------------------------------- upload

    def begin(self, kind: str, metadata: dict) -> int:
        if kind not in ("slice", "file"):
            raise UploadError(f"unknown upload kind {kind!r}")
        with self._lock:
            if self._active is not None:
                raise ParallelUploadError(
                    "another upload is in progress; parallel uploads are "
                    "not supported")
            uid = self._next_id
            self._next_id += 1
            name = metadata.get("name") or f"upload_{uid}"
            sub = "slices" if kind == "slice" else "files"
            path = os.path.join(self.root, sub, name)
            rec = UploadRecord(upload_id=uid, kind=kind, metadata=metadata,
                               path=path)
            self._records[uid] = rec
            self._active = uid
            self._fh = open(path, "wb")
            self._hasher = hashlib.sha256()
            return uid

    def part(self, upload_id: int, data: bytes) -> int:
        with self._lock:
            rec = self._get_active(upload_id)
            self._fh.write(data)
            self._hasher.update(data)
            rec.size += len(data)
            return rec.size

    def end(self, upload_id: int, total_size: int, checksum: str) -> UploadRecord:
        with self._lock:
            rec = self._get_active(upload_id)
            self._fh.close()
            digest = self._hasher.hexdigest()
            self._fh = None
            self._hasher = None
            self._active = None
            if rec.size != total_size or digest != checksum:
                rec.failed = True
                try:
                    os.unlink(rec.path)
                except OSError:
                    pass
                self._persist()
                raise FailedUploadError(
                    f"upload {upload_id} failed: got {rec.size} B "
                    f"(expected {total_size}), sha256 {digest[:12]}… "
                    f"(expected {checksum[:12]}…)")
            rec.finished = True
            rec.checksum = digest
            self._persist()
            return rec

    def abort_active(self) -> None:
        with self._lock:
            if self._active is None:
                return
            rec = self._records[self._active]
            rec.failed = True
            if self._fh:
                self._fh.close()
            try:
                os.unlink(rec.path)
            except OSError:
                pass
            self._fh = None
            self._hasher = None
            self._active = None
            self._persist()

    def _get_active(self, upload_id: int) -> UploadRecord:
        if upload_id not in self._records:
            raise UploadNotFoundError(f"no upload {upload_id}")
        if self._active != upload_id:
            raise UploadNotFoundError(f"upload {upload_id} is not active")
        return self._records[upload_id]

    # -------------------------------------------------------------- query

    def finished(self, kind: Optional[str] = None) -> List[UploadRecord]:
        with self._lock:
            return [r for r in self._records.values()
                    if r.finished and (kind is None or r.kind == kind)]

    def find_slice(self, name: str) -> Optional[UploadRecord]:
        for r in self.finished("slice"):
            if r.metadata.get("name") == name or \
                    os.path.basename(r.path) == name:
                return r
        return None
