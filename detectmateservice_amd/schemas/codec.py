"""Minimal proto3 wire-format codec (dependency-free).

The inter-stage wire format is proto3 (reference: embedded descriptor in
/root/reference/container/fluentout/schemas_pb.rb:9, decoded in SURVEY.md
§2.3). The messages are small and fixed, so this framework carries its own
codec instead of generated stubs: a declarative field spec per message plus
varint/length-delimited encode/decode helpers. A batched C++ decoder that
turns N frames into SoA tensors for the GPU stages lives in
``detectmateservice_amd/ops`` and shares this wire format.

Wire types used: 0 = varint (int32/int64/bool), 2 = length-delimited
(string/bytes/sub-message/packed repeated int), 5 = fixed32 (float).
proto3 semantics honored: default values are not emitted; packed repeated
ints are written packed and read either packed or unpacked; maps encode as
repeated embedded messages {1: key, 2: value}; unknown fields are skipped.
"""
from __future__ import annotations

import struct
from typing import Any, Dict, Iterator, List, Tuple

# ---------------------------------------------------------------------------
# varint / tag primitives
# ---------------------------------------------------------------------------


def encode_varint(value: int) -> bytes:
    """Encode a non-negative int as base-128 varint."""
    if value < 0:
        # proto3 int32 negative values are sign-extended to 64 bits
        value &= (1 << 64) - 1
    out = bytearray()
    while True:
        b = value & 0x7F
        value >>= 7
        if value:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def decode_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    """Decode a varint at ``pos``; returns (value, new_pos)."""
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
        if shift >= 70:
            raise ValueError("varint too long")


def _to_int32(v: int) -> int:
    """Interpret a decoded varint as a signed int32 (two's complement)."""
    v &= (1 << 64) - 1
    v &= 0xFFFFFFFF
    return v - (1 << 32) if v >= (1 << 31) else v


def encode_tag(field_number: int, wire_type: int) -> bytes:
    return encode_varint((field_number << 3) | wire_type)


def _skip_field(buf: bytes, pos: int, wire_type: int) -> int:
    if wire_type == 0:
        _, pos = decode_varint(buf, pos)
    elif wire_type == 1:
        pos += 8
    elif wire_type == 2:
        n, pos = decode_varint(buf, pos)
        pos += n
    elif wire_type == 5:
        pos += 4
    else:
        raise ValueError(f"unsupported wire type {wire_type}")
    if pos > len(buf):
        raise ValueError("truncated field")
    return pos


# ---------------------------------------------------------------------------
# field kinds
# ---------------------------------------------------------------------------

STRING = "string"
INT32 = "int32"
FLOAT = "float"
REP_STRING = "rep_string"
REP_INT32 = "rep_int32"
MAP_SS = "map_ss"

_DEFAULTS = {
    STRING: "",
    INT32: 0,
    FLOAT: 0.0,
    REP_STRING: list,
    REP_INT32: list,
    MAP_SS: dict,
}


def default_for(kind: str) -> Any:
    d = _DEFAULTS[kind]
    return d() if callable(d) else d


# ---------------------------------------------------------------------------
# encode / decode over a declarative spec
# ---------------------------------------------------------------------------
# spec: Dict[str field_name, Tuple[int field_number, str kind]]


def encode_message(spec: Dict[str, Tuple[int, str]], values: Dict[str, Any]) -> bytes:
    out = bytearray()
    # Emit in field-number order for deterministic bytes.
    for name, (num, kind) in sorted(spec.items(), key=lambda kv: kv[1][0]):
        v = values.get(name)
        if v is None:
            continue
        if kind == STRING:
            if v == "":
                continue
            b = v.encode("utf-8") if isinstance(v, str) else bytes(v)
            out += encode_tag(num, 2) + encode_varint(len(b)) + b
        elif kind == INT32:
            if int(v) == 0:
                continue
            out += encode_tag(num, 0) + encode_varint(int(v))
        elif kind == FLOAT:
            if float(v) == 0.0:
                continue
            out += encode_tag(num, 5) + struct.pack("<f", float(v))
        elif kind == REP_STRING:
            for item in v:
                b = item.encode("utf-8") if isinstance(item, str) else bytes(item)
                out += encode_tag(num, 2) + encode_varint(len(b)) + b
        elif kind == REP_INT32:
            if not v:
                continue
            packed = b"".join(encode_varint(int(x)) for x in v)
            out += encode_tag(num, 2) + encode_varint(len(packed)) + packed
        elif kind == MAP_SS:
            for k in v:
                kb = k.encode("utf-8")
                vb = v[k].encode("utf-8")
                entry = (
                    encode_tag(1, 2) + encode_varint(len(kb)) + kb
                    + encode_tag(2, 2) + encode_varint(len(vb)) + vb
                )
                out += encode_tag(num, 2) + encode_varint(len(entry)) + entry
        else:
            raise ValueError(f"unknown field kind {kind}")
    return bytes(out)


def decode_message(spec: Dict[str, Tuple[int, str]], data: bytes) -> Dict[str, Any]:
    by_num = {num: (name, kind) for name, (num, kind) in spec.items()}
    values: Dict[str, Any] = {}
    pos = 0
    n = len(data)
    while pos < n:
        key, pos = decode_varint(data, pos)
        num, wt = key >> 3, key & 7
        entry = by_num.get(num)
        if entry is None:
            pos = _skip_field(data, pos, wt)
            continue
        name, kind = entry
        if kind == STRING and wt == 2:
            ln, pos = decode_varint(data, pos)
            values[name] = data[pos:pos + ln].decode("utf-8", errors="replace")
            pos += ln
        elif kind == INT32 and wt == 0:
            v, pos = decode_varint(data, pos)
            values[name] = _to_int32(v)
        elif kind == FLOAT and wt == 5:
            if pos + 4 > n:
                raise ValueError("truncated fixed32 field")
            values[name] = struct.unpack_from("<f", data, pos)[0]
            pos += 4
        elif kind == REP_STRING and wt == 2:
            ln, pos = decode_varint(data, pos)
            values.setdefault(name, []).append(
                data[pos:pos + ln].decode("utf-8", errors="replace")
            )
            pos += ln
        elif kind == REP_INT32:
            if wt == 2:  # packed
                ln, pos = decode_varint(data, pos)
                end = pos + ln
                lst = values.setdefault(name, [])
                while pos < end:
                    v, pos = decode_varint(data, pos)
                    lst.append(_to_int32(v))
            elif wt == 0:  # unpacked
                v, pos = decode_varint(data, pos)
                values.setdefault(name, []).append(_to_int32(v))
            else:
                pos = _skip_field(data, pos, wt)
        elif kind == MAP_SS and wt == 2:
            ln, pos = decode_varint(data, pos)
            entry_bytes = data[pos:pos + ln]
            pos += ln
            k = v = ""
            epos = 0
            while epos < len(entry_bytes):
                ekey, epos = decode_varint(entry_bytes, epos)
                enum_, ewt = ekey >> 3, ekey & 7
                if ewt == 2:
                    eln, epos = decode_varint(entry_bytes, epos)
                    s = entry_bytes[epos:epos + eln].decode("utf-8", errors="replace")
                    epos += eln
                    if enum_ == 1:
                        k = s
                    elif enum_ == 2:
                        v = s
                else:
                    epos = _skip_field(entry_bytes, epos, ewt)
            values.setdefault(name, {})[k] = v
        else:
            pos = _skip_field(data, pos, wt)
        if pos > n:
            raise ValueError("truncated message")
    return values
