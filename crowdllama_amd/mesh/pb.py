"""Protobuf wire schema for the inference protocol.

Equivalent of the reference's external crowdllama-pb package (SURVEY.md
§2.2: BaseMessage oneof{GenerateRequest, GenerateResponse}). Hand-rolled
proto3 wire-format codec (no protoc in the image); the bytes are valid
protobuf, decodable by any proto library given the schema:

    message GenerateRequest  { string model=1; string prompt=2; bool stream=3; }
    message Timestamp        { int64 seconds=1; int32 nanos=2; }
    message GenerateResponse { string model=1; Timestamp created_at=2;
                               string response=3; bool done=4;
                               string done_reason=5; string worker_id=6;
                               int64 total_duration=7;
                               int64 eval_count=8 /*extension*/; }
    message BaseMessage      { oneof msg { GenerateRequest generate_request=1;
                                           GenerateResponse generate_response=2; } }
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field


# ---------------------------------------------------------------- varint --

def _enc_varint(v: int) -> bytes:
    if v < 0:
        v += 1 << 64
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _dec_varint(buf: bytes, pos: int) -> tuple[int, int]:
    result = 0
    shift = 0
    while True:
        if pos >= len(buf):
            raise ValueError("truncated varint")
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            return result, pos
        shift += 7
        if shift > 70:
            raise ValueError("varint too long")


def _tag(fieldno: int, wire: int) -> bytes:
    return _enc_varint((fieldno << 3) | wire)


def _enc_str(fieldno: int, s: str) -> bytes:
    b = s.encode("utf-8")
    return _tag(fieldno, 2) + _enc_varint(len(b)) + b if b else b""


def _enc_bytes(fieldno: int, b: bytes) -> bytes:
    return _tag(fieldno, 2) + _enc_varint(len(b)) + b


def _enc_int(fieldno: int, v: int) -> bytes:
    return (_tag(fieldno, 0) + _enc_varint(v)) if v else b""


def _enc_bool(fieldno: int, v: bool) -> bytes:
    return (_tag(fieldno, 0) + b"\x01") if v else b""


def _iter_fields(buf: bytes):
    pos = 0
    while pos < len(buf):
        key, pos = _dec_varint(buf, pos)
        fieldno, wire = key >> 3, key & 7
        if wire == 0:
            v, pos = _dec_varint(buf, pos)
            yield fieldno, wire, v
        elif wire == 2:
            n, pos = _dec_varint(buf, pos)
            yield fieldno, wire, buf[pos:pos + n]
            pos += n
        elif wire == 5:
            yield fieldno, wire, buf[pos:pos + 4]
            pos += 4
        elif wire == 1:
            yield fieldno, wire, buf[pos:pos + 8]
            pos += 8
        else:
            raise ValueError(f"unsupported wire type {wire}")


def _signed(v: int) -> int:
    return v - (1 << 64) if v >= (1 << 63) else v


# --------------------------------------------------------------- messages --

@dataclass
class GenerateRequest:
    model: str = ""
    prompt: str = ""
    stream: bool = False

    def encode(self) -> bytes:
        return (_enc_str(1, self.model) + _enc_str(2, self.prompt)
                + _enc_bool(3, self.stream))

    @classmethod
    def decode(cls, buf: bytes) -> "GenerateRequest":
        m = cls()
        for fn, wire, v in _iter_fields(buf):
            if fn == 1 and wire == 2:
                m.model = v.decode("utf-8")
            elif fn == 2 and wire == 2:
                m.prompt = v.decode("utf-8")
            elif fn == 3 and wire == 0:
                m.stream = bool(v)
        return m


@dataclass
class Timestamp:
    seconds: int = 0
    nanos: int = 0

    @classmethod
    def now(cls) -> "Timestamp":
        t = time.time()
        return cls(int(t), int((t % 1) * 1e9))

    def encode(self) -> bytes:
        return _enc_int(1, self.seconds) + _enc_int(2, self.nanos)

    @classmethod
    def decode(cls, buf: bytes) -> "Timestamp":
        m = cls()
        for fn, wire, v in _iter_fields(buf):
            if fn == 1 and wire == 0:
                m.seconds = _signed(v)
            elif fn == 2 and wire == 0:
                m.nanos = _signed(v)
        return m


@dataclass
class GenerateResponse:
    model: str = ""
    created_at: Timestamp = field(default_factory=Timestamp)
    response: str = ""
    done: bool = True
    done_reason: str = ""
    worker_id: str = ""
    total_duration: int = 0  # nanoseconds, reference parity (api.go:84)
    eval_count: int = 0      # generated tokens (extension; Ollama exposes
    #                          the same field name in its chat responses)

    def encode(self) -> bytes:
        return (_enc_str(1, self.model)
                + _enc_bytes(2, self.created_at.encode())
                + _enc_str(3, self.response)
                + _enc_bool(4, self.done)
                + _enc_str(5, self.done_reason)
                + _enc_str(6, self.worker_id)
                + _enc_int(7, self.total_duration)
                + _enc_int(8, self.eval_count))

    @classmethod
    def decode(cls, buf: bytes) -> "GenerateResponse":
        m = cls(done=False)
        for fn, wire, v in _iter_fields(buf):
            if fn == 1 and wire == 2:
                m.model = v.decode("utf-8")
            elif fn == 2 and wire == 2:
                m.created_at = Timestamp.decode(v)
            elif fn == 3 and wire == 2:
                m.response = v.decode("utf-8")
            elif fn == 4 and wire == 0:
                m.done = bool(v)
            elif fn == 5 and wire == 2:
                m.done_reason = v.decode("utf-8")
            elif fn == 6 and wire == 2:
                m.worker_id = v.decode("utf-8")
            elif fn == 7 and wire == 0:
                m.total_duration = _signed(v)
            elif fn == 8 and wire == 0:
                m.eval_count = _signed(v)
        return m


@dataclass
class BaseMessage:
    """oneof envelope (reference BaseMessage)."""
    generate_request: GenerateRequest | None = None
    generate_response: GenerateResponse | None = None

    def encode(self) -> bytes:
        if self.generate_request is not None:
            return _enc_bytes(1, self.generate_request.encode())
        if self.generate_response is not None:
            return _enc_bytes(2, self.generate_response.encode())
        return b""

    @classmethod
    def decode(cls, buf: bytes) -> "BaseMessage":
        m = cls()
        for fn, wire, v in _iter_fields(buf):
            if fn == 1 and wire == 2:
                m.generate_request = GenerateRequest.decode(v)
                m.generate_response = None
            elif fn == 2 and wire == 2:
                m.generate_response = GenerateResponse.decode(v)
                m.generate_request = None
        return m


def request_message(model: str, prompt: str, stream: bool = False) -> BaseMessage:
    return BaseMessage(generate_request=GenerateRequest(model, prompt, stream))


def response_message(model: str, response: str, worker_id: str = "",
                     done_reason: str = "stop",
                     total_duration_ns: int = 0,
                     done: bool = True, eval_count: int = 0) -> BaseMessage:
    # done=False frames are streamed chunks (capability extension over the
    # reference, which carries `stream` but never streams — SURVEY.md §2.2)
    return BaseMessage(generate_response=GenerateResponse(
        model=model, created_at=Timestamp.now(), response=response, done=done,
        done_reason=done_reason if done else "", worker_id=worker_id,
        total_duration=total_duration_ns, eval_count=eval_count))
