"""Hand-written proto3 wire codec for indexer.v1 (api/indexer.proto).

grpc_tools is not available in this image, so the three messages are
encoded/decoded directly against the proto3 wire format - byte-compatible
with any generated stub speaking the reference's api/indexer.proto:24-43:

    service IndexerService { rpc GetPodScores(GetPodScoresRequest)
                             returns (GetPodScoresResponse); }
    GetPodScoresRequest  { string prompt = 1; string model_name = 2;
                           repeated string pod_identifiers = 3; }
    GetPodScoresResponse { repeated PodScore scores = 1; }
    PodScore             { string pod = 1; double score = 2; }
"""

from __future__ import annotations

import struct
from dataclasses import dataclass, field
from typing import List, Tuple

SERVICE_NAME = "indexer.v1.IndexerService"
GET_POD_SCORES_METHOD = f"/{SERVICE_NAME}/GetPodScores"


def _encode_varint(v: int) -> bytes:
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _decode_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        if pos >= len(buf):
            raise ValueError("truncated varint")
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7
        if shift > 63:
            raise ValueError("varint too long")


def _tag(field_no: int, wire_type: int) -> bytes:
    return _encode_varint((field_no << 3) | wire_type)


def _len_delim(field_no: int, payload: bytes) -> bytes:
    return _tag(field_no, 2) + _encode_varint(len(payload)) + payload


def _iter_fields(buf: bytes):
    pos = 0
    while pos < len(buf):
        key, pos = _decode_varint(buf, pos)
        field_no, wire_type = key >> 3, key & 7
        if wire_type == 0:
            val, pos = _decode_varint(buf, pos)
        elif wire_type == 1:
            val = buf[pos : pos + 8]
            pos += 8
        elif wire_type == 2:
            ln, pos = _decode_varint(buf, pos)
            val = buf[pos : pos + ln]
            pos += ln
        elif wire_type == 5:
            val = buf[pos : pos + 4]
            pos += 4
        else:
            raise ValueError(f"unsupported wire type {wire_type}")
        yield field_no, wire_type, val


@dataclass
class GetPodScoresRequest:
    prompt: str = ""
    model_name: str = ""
    pod_identifiers: List[str] = field(default_factory=list)

    def encode(self) -> bytes:
        out = bytearray()
        if self.prompt:
            out += _len_delim(1, self.prompt.encode("utf-8"))
        if self.model_name:
            out += _len_delim(2, self.model_name.encode("utf-8"))
        for pod in self.pod_identifiers:
            out += _len_delim(3, pod.encode("utf-8"))
        return bytes(out)

    @staticmethod
    def decode(buf: bytes) -> "GetPodScoresRequest":
        req = GetPodScoresRequest()
        for field_no, wt, val in _iter_fields(buf):
            if field_no == 1 and wt == 2:
                req.prompt = val.decode("utf-8")
            elif field_no == 2 and wt == 2:
                req.model_name = val.decode("utf-8")
            elif field_no == 3 and wt == 2:
                req.pod_identifiers.append(val.decode("utf-8"))
        return req


@dataclass
class PodScore:
    pod: str = ""
    score: float = 0.0

    def encode(self) -> bytes:
        out = bytearray()
        if self.pod:
            out += _len_delim(1, self.pod.encode("utf-8"))
        if self.score != 0.0:
            out += _tag(2, 1) + struct.pack("<d", self.score)
        return bytes(out)

    @staticmethod
    def decode(buf: bytes) -> "PodScore":
        ps = PodScore()
        for field_no, wt, val in _iter_fields(buf):
            if field_no == 1 and wt == 2:
                ps.pod = val.decode("utf-8")
            elif field_no == 2 and wt == 1:
                ps.score = struct.unpack("<d", val)[0]
        return ps


@dataclass
class GetPodScoresResponse:
    scores: List[PodScore] = field(default_factory=list)

    def encode(self) -> bytes:
        out = bytearray()
        for s in self.scores:
            out += _len_delim(1, s.encode())
        return bytes(out)

    @staticmethod
    def decode(buf: bytes) -> "GetPodScoresResponse":
        resp = GetPodScoresResponse()
        for field_no, wt, val in _iter_fields(buf):
            if field_no == 1 and wt == 2:
                resp.scores.append(PodScore.decode(val))
        return resp
