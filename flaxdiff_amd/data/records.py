"""Record-shard dataset format — the local ArrayRecord equivalent.

The reference trains from ArrayRecord shards on GCS via grain
(/root/reference/flaxdiff/data/dataset_map.py:19-174, sources/gcs.py); this
is the MI355X-native stand-in: a length-prefixed msgpack record file with an
offset index for O(1) random access, shardable across DP ranks and loader
workers by the existing ShardedSampler. No network, no TF deps.

Format per shard (`<name>.rec`):
    repeat: [u64 little-endian payload length][msgpack payload]
Index (`<name>.rec.idx`): u64 array of record byte offsets.

ndarrays are encoded losslessly as {"__nd__": 1, dtype, shape, data-bytes};
bytes/str/int/float/list/dict pass through msgpack untouched.
"""
from __future__ import annotations

import glob
import os
import struct
from typing import Any, Dict, Iterable, List, Optional, Sequence

import numpy as np

try:
    import msgpack
except Exception:  # pragma: no cover
    msgpack = None

from .sources import DataSource


def _require_msgpack():
    if msgpack is None:
        raise RuntimeError("record shards need the msgpack package")
    return msgpack


def _enc(obj: Any):
    if isinstance(obj, np.ndarray):
        return {"__nd__": 1, "dtype": str(obj.dtype),
                "shape": list(obj.shape),
                "data": np.ascontiguousarray(obj).tobytes()}
    return obj


def _dec(obj: Any):
    if isinstance(obj, dict) and obj.get("__nd__") == 1:
        arr = np.frombuffer(obj["data"], dtype=np.dtype(obj["dtype"]))
        return arr.reshape(obj["shape"]).copy()
    return obj


class RecordShardWriter:
    """Append-only writer for one `.rec` shard (+ its offset index)."""

    def __init__(self, path: str):
        _require_msgpack()
        if not path.endswith(".rec"):
            path += ".rec"
        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        self.path = path
        self._f = open(path, "wb")
        self._offsets: List[int] = []

    def write(self, record: Dict[str, Any]):
        payload = msgpack.packb({k: _enc(v) for k, v in record.items()},
                                use_bin_type=True)
        self._offsets.append(self._f.tell())
        self._f.write(struct.pack("<Q", len(payload)))
        self._f.write(payload)

    def __len__(self):
        return len(self._offsets)

    def close(self):
        if self._f is None:
            return
        self._f.close()
        self._f = None
        np.asarray(self._offsets, dtype=np.uint64).tofile(self.path + ".idx")

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()


class RecordShardDataset:
    """Random-access reader over one or many `.rec` shards.

    `paths` may be a glob, a directory, a single shard, or a list. File
    handles are opened lazily PER PROCESS (safe with DataLoader workers:
    each worker re-opens on first access after fork/spawn).
    """

    def __init__(self, paths):
        _require_msgpack()
        if isinstance(paths, (str, os.PathLike)):
            p = str(paths)
            if os.path.isdir(p):
                paths = sorted(glob.glob(os.path.join(p, "*.rec")))
            elif any(ch in p for ch in "*?["):
                paths = sorted(glob.glob(p))
            else:
                paths = [p]
        self.paths = [str(p) for p in paths]
        if not self.paths:
            raise FileNotFoundError("no .rec shards found")
        self._index = []          # (path_idx, offset)
        self._counts = []
        for pi, path in enumerate(self.paths):
            offs = np.fromfile(path + ".idx", dtype=np.uint64)
            self._counts.append(len(offs))
            self._index.append(offs)
        self._cum = np.cumsum([0] + self._counts)
        self._handles: Dict[int, Any] = {}
        self._pid = None

    def __len__(self):
        return int(self._cum[-1])

    def _handle(self, pi: int):
        if self._pid != os.getpid():
            self._handles = {}
            self._pid = os.getpid()
        h = self._handles.get(pi)
        if h is None:
            h = open(self.paths[pi], "rb")
            self._handles[pi] = h
        return h

    def __getitem__(self, i: int) -> Dict[str, Any]:
        if i < 0:
            i += len(self)
        pi = int(np.searchsorted(self._cum, i, side="right") - 1)
        off = int(self._index[pi][i - self._cum[pi]])
        f = self._handle(pi)
        f.seek(off)
        (n,) = struct.unpack("<Q", f.read(8))
        rec = msgpack.unpackb(f.read(n), raw=False)
        return {k: _dec(v) for k, v in rec.items()}

    def __getstate__(self):
        d = dict(self.__dict__)
        d["_handles"] = {}
        d["_pid"] = None
        return d


class RecordSource(DataSource):
    """DataSource over record shards (plugs into the dataset registry and
    the sharded/prefetching loaders like every other source)."""

    def __init__(self, paths, image_key: str = "image",
                 caption_key: str = "caption"):
        self.ds = RecordShardDataset(paths)
        self.image_key = image_key
        self.caption_key = caption_key

    def __len__(self):
        return len(self.ds)

    def __getitem__(self, idx: int) -> Dict[str, Any]:
        rec = self.ds[idx]
        out = dict(rec)
        if self.image_key != "image" and self.image_key in rec:
            out["image"] = rec[self.image_key]
        if self.caption_key != "caption" and self.caption_key in rec:
            out["caption"] = rec[self.caption_key]
        return out


def register_record_dataset(name: str, paths, image_size: int = 64):
    """Register a record-shard dataset under the named-dataset registry
    (the reference's dataset_map arrayrecord entries, dataset_map.py:19-61)."""
    from .sources import ImageAugmenter, register_dataset

    def build(**kwargs):
        return (RecordSource(paths),
                ImageAugmenter(image_size=image_size,
                               tokenizer=kwargs.get("tokenizer")))
    register_dataset(name, build)


def write_records_from_source(source: DataSource, out_dir: str,
                              shard_size: int = 1024,
                              prefix: str = "data") -> List[str]:
    """Convert any DataSource into record shards (the role of the
    reference's img2dataset / convert_hf_to_arrayrecord jobs)."""
    os.makedirs(out_dir, exist_ok=True)
    paths = []
    w = None
    for i in range(len(source)):
        if w is None or len(w) >= shard_size:
            if w is not None:
                w.close()
            path = os.path.join(out_dir, f"{prefix}-{len(paths):05d}.rec")
            paths.append(path)
            w = RecordShardWriter(path)
        w.write(source[i])
    if w is not None:
        w.close()
    return paths
