"""Batch construction: the user-facing data model.

Mirrors reference ``persia/embedding/data.py`` (same class names, same
validation semantics, same ``MAX_BATCH_SIZE = 65535`` u16 sample-coordinate
limit — persia/embedding/data.py:14, persia-common/src/lib.rs:49-51), but the
native representation is a flat CSR (values + sample offsets) per slot, built
once on the host and uploaded to the GPU as-is — there is no per-sign CPU
hashmap dedup here (reference FeatureBatch::new, persia-common/src/lib.rs:46-82);
dedup happens on-GPU via radix sort.
"""
import struct
from typing import List, Optional, Union

import numpy as np

from persia_amd.env import PERSIA_SKIP_CHECK_DATA

MAX_BATCH_SIZE = 65535

_ND_ARRAY_SUPPORT_TYPE = {
    np.bool_,
    np.int8,
    np.int16,
    np.int32,
    np.int64,
    np.float32,
    np.float64,
    np.uint8,
}

_DTYPE_CODE = {
    np.dtype(np.bool_): 0,
    np.dtype(np.int8): 1,
    np.dtype(np.int16): 2,
    np.dtype(np.int32): 3,
    np.dtype(np.int64): 4,
    np.dtype(np.float32): 5,
    np.dtype(np.float64): 6,
    np.dtype(np.uint8): 7,
    np.dtype(np.uint64): 8,
}
_CODE_DTYPE = {v: k for k, v in _DTYPE_CODE.items()}


def _id_type_data_check(data: np.ndarray, name: str) -> None:
    """Same validation semantics as the reference (persia/embedding/data.py:
    21-40): per-sample ID lists must be 1-D uint64 numpy arrays."""
    assert isinstance(data, np.ndarray), (
        f"id_type_feature {name!r}: each sample must be a numpy ndarray, "
        f"not {type(data).__name__}"
    )
    assert data.ndim == 1, (
        f"id_type_feature {name!r}: samples are 1-D ID lists, got a "
        f"{data.ndim}-D array"
    )
    assert data.dtype == np.uint64, (
        f"id_type_feature {name!r}: IDs must be uint64 (sign space), got "
        f"{data.dtype}"
    )


def _ndarray_check(data: np.ndarray, name: str) -> None:
    """Dense-tensor validation (reference persia/embedding/data.py:41-56)."""
    assert isinstance(data, np.ndarray), (
        f"{name!r}: dense data must be a numpy ndarray, not "
        f"{type(data).__name__}"
    )
    assert data.dtype.type in _ND_ARRAY_SUPPORT_TYPE, (
        f"{name!r}: dtype {data.dtype} is not wire-serializable; supported: "
        f"{sorted(t.__name__ for t in _ND_ARRAY_SUPPORT_TYPE)}"
    )
    assert data.ndim > 0, f"{name!r}: dense data cannot be 0-dimensional"


def _batch_size_check(batch_size: int, target: int, data_type: str, name: str) -> None:
    """Every component of a PersiaBatch must agree on the sample count, which
    is capped by the u16 wire coordinate (reference data.py:57-68)."""
    assert batch_size == target, (
        f"{data_type} {name!r}: has {batch_size} samples but the batch has "
        f"{target}"
    )
    assert batch_size <= MAX_BATCH_SIZE, (
        f"{data_type} {name!r}: {batch_size} samples exceeds the "
        f"u16-coordinate cap of {MAX_BATCH_SIZE}"
    )


class IDTypeFeature:
    """A LIL sparse matrix of categorical uint64 IDs, one variable-length row
    per sample (mirrors persia/embedding/data.py:69-113)."""

    def __init__(self, name: str, data: List[np.ndarray]):
        if not PERSIA_SKIP_CHECK_DATA:
            for x in data:
                _id_type_data_check(x, name)
        self.name = name
        self.data = data

    @property
    def batch_size(self) -> int:
        return len(self.data)

    def flatten(self):
        """-> (values u64[nnz], offsets i64[batch+1])"""
        lens = np.fromiter((len(x) for x in self.data), dtype=np.int64, count=len(self.data))
        offsets = np.zeros(len(self.data) + 1, dtype=np.int64)
        np.cumsum(lens, out=offsets[1:])
        if offsets[-1] == 0:
            values = np.empty(0, dtype=np.uint64)
        else:
            values = np.concatenate([np.asarray(x, dtype=np.uint64) for x in self.data])
        return values, offsets


class IDTypeFeatureWithSingleID:
    """Exactly one ID per sample (mirrors persia/embedding/data.py:116-157);
    avoids the per-sample array overhead of :class:`IDTypeFeature`."""

    def __init__(self, name: str, data: np.ndarray):
        if not PERSIA_SKIP_CHECK_DATA:
            _id_type_data_check(data, name)
        self.name = name
        self.data = data

    @property
    def batch_size(self) -> int:
        return len(self.data)

    def flatten(self):
        n = len(self.data)
        return np.asarray(self.data, dtype=np.uint64), np.arange(n + 1, dtype=np.int64)


class NdarrayDataBase:
    DEFAULT_NAME = "ndarray_base"

    def __init__(self, data: np.ndarray, name: Optional[str] = None):
        if not PERSIA_SKIP_CHECK_DATA:
            _ndarray_check(data, name or self.DEFAULT_NAME)
        self.data = np.ascontiguousarray(data)
        self.name = name or self.DEFAULT_NAME

    @property
    def batch_size(self) -> int:
        return self.data.shape[0]


class NonIDTypeFeature(NdarrayDataBase):
    DEFAULT_NAME = "non_id_type_feature"


class Label(NdarrayDataBase):
    DEFAULT_NAME = "label"


class _FlatIDFeature:
    """Internal flat CSR form of one slot's ID batch.

    ``is_single`` marks exactly-one-id-per-sample slots (offsets == arange):
    the engine's static-plan fast path applies to batches of only such slots.
    """

    __slots__ = ("name", "values", "offsets", "is_single")

    def __init__(self, name: str, values: np.ndarray, offsets: np.ndarray,
                 is_single: bool = False):
        self.name = name
        self.values = values
        self.offsets = offsets
        self.is_single = is_single

    @property
    def batch_size(self) -> int:
        return len(self.offsets) - 1


_MAGIC = b"PAB1"


class PersiaBatch:
    """One training/inference batch (mirrors persia/embedding/data.py:279-411).

    Arguments:
        id_type_features: list of :class:`IDTypeFeature` /
            :class:`IDTypeFeatureWithSingleID` (at least one required).
        non_id_type_features: optional list of :class:`NonIDTypeFeature` or
            raw ndarrays.
        labels: optional list of :class:`Label` or raw ndarrays
            (required when ``requires_grad``).
        requires_grad: whether this batch takes the training path.
        meta: optional user bytes carried alongside.
    """

    def __init__(
        self,
        id_type_features: List[Union[IDTypeFeature, IDTypeFeatureWithSingleID]],
        non_id_type_features: Optional[List[Union[NonIDTypeFeature, np.ndarray]]] = None,
        labels: Optional[List[Union[Label, np.ndarray]]] = None,
        batch_size: Optional[int] = None,
        requires_grad: bool = True,
        meta: Optional[bytes] = None,
    ):
        assert len(id_type_features) > 0, "id_type_features should not be empty"
        batch_size = batch_size or id_type_features[0].batch_size
        assert batch_size <= MAX_BATCH_SIZE, (
            f"expected batch_size <= MAX_BATCH_SIZE: {MAX_BATCH_SIZE} but got {batch_size}"
        )

        self.id_type_features: List[_FlatIDFeature] = []
        seen = set()
        for f in id_type_features:
            if not PERSIA_SKIP_CHECK_DATA:
                _batch_size_check(f.batch_size, batch_size, "id_type_feature", f.name)
            assert f.name not in seen, f"duplicate id_type_feature name: {f.name}"
            seen.add(f.name)
            values, offsets = f.flatten()
            self.id_type_features.append(
                _FlatIDFeature(
                    f.name, values, offsets,
                    is_single=isinstance(f, IDTypeFeatureWithSingleID),
                )
            )

        self.non_id_type_features: List[NonIDTypeFeature] = []
        for i, x in enumerate(non_id_type_features or []):
            if isinstance(x, np.ndarray):
                x = NonIDTypeFeature(x, name=f"non_id_type_feature_{i}")
            if not PERSIA_SKIP_CHECK_DATA:
                _batch_size_check(x.batch_size, batch_size, "non_id_type_feature", x.name)
            self.non_id_type_features.append(x)

        self.labels: List[Label] = []
        for i, x in enumerate(labels or []):
            if isinstance(x, np.ndarray):
                x = Label(x, name=f"label_{i}")
            if not PERSIA_SKIP_CHECK_DATA:
                _batch_size_check(x.batch_size, batch_size, "label", x.name)
            self.labels.append(x)

        if requires_grad:
            assert len(self.labels) > 0, "labels should not be empty when requires_grad=True"

        self.batch_size = batch_size
        self.requires_grad = requires_grad
        self.meta = meta
        self.batch_id: Optional[int] = None

    # ---- serialization (replaces the reference's speedy wire format with a
    # documented little-endian layout; used by DataCtx -> trainer transport).

    def to_bytes(self) -> bytes:
        parts = [_MAGIC]

        def put_arr(a: np.ndarray):
            a = np.ascontiguousarray(a)
            code = _DTYPE_CODE[a.dtype]
            parts.append(struct.pack("<BB", code, a.ndim))
            parts.append(struct.pack(f"<{a.ndim}q", *a.shape))
            parts.append(a.tobytes())

        def put_str(s: str):
            b = s.encode("utf-8")
            parts.append(struct.pack("<I", len(b)))
            parts.append(b)

        parts.append(
            struct.pack(
                "<IB q I I I",
                self.batch_size,
                1 if self.requires_grad else 0,
                -1 if self.batch_id is None else self.batch_id,
                len(self.id_type_features),
                len(self.non_id_type_features),
                len(self.labels),
            )
        )
        for f in self.id_type_features:
            put_str(f.name)
            put_arr(f.values)
            put_arr(f.offsets)
        for x in self.non_id_type_features:
            put_str(x.name)
            put_arr(x.data)
        for x in self.labels:
            put_str(x.name)
            put_arr(x.data)
        meta = self.meta or b""
        parts.append(struct.pack("<I", len(meta)))
        parts.append(meta)
        return b"".join(parts)

    @staticmethod
    def from_bytes(buf: bytes) -> "PersiaBatch":
        assert buf[:4] == _MAGIC, "bad PersiaBatch magic"
        off = [4]

        def take(n):
            s = buf[off[0] : off[0] + n]
            off[0] += n
            return s

        def get_arr() -> np.ndarray:
            code, ndim = struct.unpack("<BB", take(2))
            shape = struct.unpack(f"<{ndim}q", take(8 * ndim))
            dt = _CODE_DTYPE[code]
            n = int(np.prod(shape)) if ndim else 1
            a = np.frombuffer(take(n * dt.itemsize), dtype=dt).reshape(shape)
            return a.copy()

        def get_str() -> str:
            (n,) = struct.unpack("<I", take(4))
            return take(n).decode("utf-8")

        batch_size, rg, batch_id, n_id, n_nid, n_lab = struct.unpack("<IB q I I I", take(25))
        obj = PersiaBatch.__new__(PersiaBatch)
        obj.batch_size = batch_size
        obj.requires_grad = bool(rg)
        obj.batch_id = None if batch_id < 0 else batch_id
        obj.id_type_features = []
        for _ in range(n_id):
            name = get_str()
            values = get_arr()
            offsets = get_arr()
            obj.id_type_features.append(
                _FlatIDFeature(
                    name, values, offsets,
                    is_single=bool(np.all(np.diff(offsets) == 1)),
                )
            )
        obj.non_id_type_features = []
        for _ in range(n_nid):
            name = get_str()
            obj.non_id_type_features.append(NonIDTypeFeature(get_arr(), name=name))
        obj.labels = []
        for _ in range(n_lab):
            name = get_str()
            obj.labels.append(Label(get_arr(), name=name))
        (n_meta,) = struct.unpack("<I", take(4))
        obj.meta = bytes(take(n_meta)) if n_meta else None
        return obj
