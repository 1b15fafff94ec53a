"""GGUF v3 container reader/writer (pure Python, numpy-backed).

The north star keeps the reference ecosystem's GGUF checkpoint format (the
reference's models are Ollama-managed GGUF files on the worker's disk —
SURVEY.md §5.4). This module provides:

- GGUFWriter: used by models/synth.py to emit random-init checkpoints for
  benches and tests.
- GGUFReader: metadata + tensor table parsing with mmap'd zero-copy tensor
  access. The C++ engine has its own mmap parser (ops/csrc/gguf.cpp); this
  one backs the CPU reference path and the tokenizer.
"""

from __future__ import annotations

import mmap
import struct
from dataclasses import dataclass
from typing import Any, BinaryIO

import numpy as np

from .kquants import GGMLType, row_bytes

GGUF_MAGIC = 0x46554747  # "GGUF" little-endian
GGUF_VERSION = 3
DEFAULT_ALIGNMENT = 32


class GGUFValueType:
    UINT8 = 0; INT8 = 1; UINT16 = 2; INT16 = 3; UINT32 = 4; INT32 = 5
    FLOAT32 = 6; BOOL = 7; STRING = 8; ARRAY = 9; UINT64 = 10; INT64 = 11
    FLOAT64 = 12


_SCALAR_FMT = {
    GGUFValueType.UINT8: "<B", GGUFValueType.INT8: "<b",
    GGUFValueType.UINT16: "<H", GGUFValueType.INT16: "<h",
    GGUFValueType.UINT32: "<I", GGUFValueType.INT32: "<i",
    GGUFValueType.FLOAT32: "<f", GGUFValueType.UINT64: "<Q",
    GGUFValueType.INT64: "<q", GGUFValueType.FLOAT64: "<d",
}


def _value_type_of(v: Any) -> int:
    if isinstance(v, bool):
        return GGUFValueType.BOOL
    if isinstance(v, int):
        return GGUFValueType.INT64 if v < 0 else (
            GGUFValueType.UINT32 if v < 2**32 else GGUFValueType.UINT64)
    if isinstance(v, float):
        return GGUFValueType.FLOAT32
    if isinstance(v, str):
        return GGUFValueType.STRING
    if isinstance(v, (list, tuple, np.ndarray)):
        return GGUFValueType.ARRAY
    raise TypeError(f"cannot encode {type(v)} in GGUF metadata")


@dataclass
class GGUFTensorInfo:
    name: str
    shape: tuple[int, ...]   # logical numpy shape (row-major, outer first)
    ggml_type: GGMLType
    offset: int              # from data-section start
    nbytes: int

    @property
    def ne(self) -> tuple[int, ...]:
        """GGUF dims (innermost first) as stored on disk."""
        return tuple(reversed(self.shape))


class GGUFWriter:
    def __init__(self, path: str):
        self.path = path
        self.kv: list[tuple[str, Any]] = []
        self.tensors: list[tuple[str, tuple[int, ...], GGMLType, np.ndarray]] = []

    def add(self, key: str, value: Any) -> None:
        self.kv.append((key, value))

    def add_tensor(self, name: str, shape: tuple[int, ...], ggml_type: GGMLType,
                   raw: np.ndarray) -> None:
        """raw: uint8 bytes of the already-encoded tensor."""
        raw = np.ascontiguousarray(raw).view(np.uint8).reshape(-1)
        expect = row_bytes(ggml_type, shape[-1]) * int(np.prod(shape[:-1], dtype=np.int64)) \
            if len(shape) > 1 else row_bytes(ggml_type, shape[-1])
        assert raw.nbytes == expect, f"{name}: {raw.nbytes} != {expect}"
        self.tensors.append((name, tuple(shape), GGMLType(ggml_type), raw))

    # --- encoding helpers ---
    @staticmethod
    def _enc_str(s: str) -> bytes:
        b = s.encode("utf-8")
        return struct.pack("<Q", len(b)) + b

    @classmethod
    def _enc_value(cls, v: Any, vt: int | None = None) -> bytes:
        if vt is None:
            vt = _value_type_of(v)
        if vt == GGUFValueType.STRING:
            return cls._enc_str(v)
        if vt == GGUFValueType.BOOL:
            return struct.pack("<B", 1 if v else 0)
        if vt == GGUFValueType.ARRAY:
            seq = list(v)
            if len(seq) == 0:
                et = GGUFValueType.UINT32
            else:
                et = _value_type_of(seq[0])
            out = struct.pack("<IQ", et, len(seq))
            for item in seq:
                out += cls._enc_value(item, et)
            return out
        return struct.pack(_SCALAR_FMT[vt], v)

    def write(self) -> None:
        align = DEFAULT_ALIGNMENT
        kv = list(self.kv)
        if not any(k == "general.alignment" for k, _ in kv):
            kv.append(("general.alignment", align))
        header = struct.pack("<IIQQ", GGUF_MAGIC, GGUF_VERSION,
                             len(self.tensors), len(kv))
        meta = b""
        for k, v in kv:
            vt = _value_type_of(v)
            meta += self._enc_str(k) + struct.pack("<I", vt) + self._enc_value(v, vt)
        # tensor infos
        infos = b""
        offset = 0
        offsets = []
        for name, shape, t, raw in self.tensors:
            offset = (offset + align - 1) // align * align
            offsets.append(offset)
            ne = tuple(reversed(shape))
            infos += self._enc_str(name)
            infos += struct.pack("<I", len(ne))
            for d in ne:
                infos += struct.pack("<Q", d)
            infos += struct.pack("<IQ", int(t), offset)
            offset += raw.nbytes
        head = header + meta + infos
        data_start = (len(head) + align - 1) // align * align
        with open(self.path, "wb") as f:
            f.write(head)
            f.write(b"\x00" * (data_start - len(head)))
            pos = 0
            for (name, shape, t, raw), off in zip(self.tensors, offsets):
                if off > pos:
                    f.write(b"\x00" * (off - pos))
                    pos = off
                f.write(memoryview(raw))
                pos += raw.nbytes


class GGUFReader:
    """mmap-backed reader. tensor_data() returns zero-copy uint8 views."""

    def __init__(self, path: str):
        self.path = path
        self._f: BinaryIO = open(path, "rb")
        self._mm = mmap.mmap(self._f.fileno(), 0, access=mmap.ACCESS_READ)
        self._pos = 0
        self.metadata: dict[str, Any] = {}
        self.tensors: dict[str, GGUFTensorInfo] = {}
        self._parse()

    def close(self) -> None:
        self._mm.close()
        self._f.close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()

    # --- decoding helpers ---
    def _read(self, n: int) -> bytes:
        b = self._mm[self._pos:self._pos + n]
        if len(b) != n:
            raise EOFError("truncated GGUF file")
        self._pos += n
        return b

    def _unpack(self, fmt: str):
        size = struct.calcsize(fmt)
        return struct.unpack(fmt, self._read(size))[0]

    def _read_str(self) -> str:
        n = self._unpack("<Q")
        return self._read(n).decode("utf-8")

    def _read_value(self, vt: int) -> Any:
        if vt == GGUFValueType.STRING:
            return self._read_str()
        if vt == GGUFValueType.BOOL:
            return self._unpack("<B") != 0
        if vt == GGUFValueType.ARRAY:
            et = self._unpack("<I")
            n = self._unpack("<Q")
            return [self._read_value(et) for _ in range(n)]
        return self._unpack(_SCALAR_FMT[vt])

    def _parse(self) -> None:
        magic = self._unpack("<I")
        if magic != GGUF_MAGIC:
            raise ValueError(f"not a GGUF file: magic {magic:#x}")
        version = self._unpack("<I")
        if version not in (2, 3):
            raise ValueError(f"unsupported GGUF version {version}")
        n_tensors = self._unpack("<Q")
        n_kv = self._unpack("<Q")
        for _ in range(n_kv):
            key = self._read_str()
            vt = self._unpack("<I")
            self.metadata[key] = self._read_value(vt)
        align = int(self.metadata.get("general.alignment", DEFAULT_ALIGNMENT))
        infos = []
        for _ in range(n_tensors):
            name = self._read_str()
            n_dims = self._unpack("<I")
            ne = [self._unpack("<Q") for _ in range(n_dims)]
            t = GGMLType(self._unpack("<I"))
            off = self._unpack("<Q")
            shape = tuple(reversed([int(d) for d in ne]))
            n_rows = 1
            for d in shape[:-1]:
                n_rows *= d
            nbytes = row_bytes(t, shape[-1]) * n_rows
            infos.append(GGUFTensorInfo(name, shape, t, off, nbytes))
        self.data_start = (self._pos + align - 1) // align * align
        for ti in infos:
            self.tensors[ti.name] = ti

    def tensor_data(self, name: str) -> np.ndarray:
        """Zero-copy uint8 view of the raw encoded tensor bytes."""
        ti = self.tensors[name]
        start = self.data_start + ti.offset
        return np.frombuffer(self._mm, dtype=np.uint8,
                             count=ti.nbytes, offset=start)

    def tensor_f32(self, name: str) -> np.ndarray:
        """Dequantized float32 tensor in its logical shape."""
        from .kquants import dequantize
        ti = self.tensors[name]
        raw = self.tensor_data(name)
        k = ti.shape[-1]
        n_rows = ti.nbytes // row_bytes(ti.ggml_type, k)
        per_row = row_bytes(ti.ggml_type, k)
        out = dequantize(raw.reshape(n_rows, per_row), ti.ggml_type, k)
        return out.reshape(ti.shape)
