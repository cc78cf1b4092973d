"""Hand-written protobuf wire codec for the reference's gRPC storage API.

No protoc is available in the build environment, so rather than generating
stubs this module encodes/decodes the exact wire format of
``optuna/storages/_grpc/api.proto`` (field numbers and types transcribed from
that file; verified byte-for-byte against the reference's generated classes by
``tests/test_grpc_wire.py``). Covered wire features: varint (int64 / bool /
enum), fixed64 (double), length-delimited (string / sub-message), proto3
packed repeated scalars, and map fields (repeated {key=1, value=2} entries).
"""
from __future__ import annotations

import struct
from typing import Any


# ---- low-level wire primitives ----------------------------------------------------


def _write_varint(out: bytearray, value: int) -> None:
    if value < 0:
        value &= (1 << 64) - 1  # two's complement, 64-bit
    while True:
        b = value & 0x7F
        value >>= 7
        if value:
            out.append(b | 0x80)
        else:
            out.append(b)
            return


def _read_varint(data: bytes, i: int) -> tuple[int, int]:
    shift = 0
    result = 0
    while True:
        b = data[i]
        i += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            break
        shift += 7
    if result >= 1 << 63:  # negative int64
        result -= 1 << 64
    return result, i


def _tag(field_number: int, wire_type: int) -> int:
    return (field_number << 3) | wire_type


# Wire types
_VARINT = 0
_FIXED64 = 1
_LEN = 2


# ---- schema-driven message codec ---------------------------------------------------
#
# A schema is {field_name: (field_number, kind)} where kind is one of:
#   "int64" | "bool" | "enum"            → varint
#   "double"                             → fixed64
#   "string"                             → length-delimited utf-8
#   ("msg", schema)                      → length-delimited sub-message
#   ("rep_int64"|"rep_double"|"rep_enum")→ packed repeated scalar
#   ("rep_msg", schema)                  → repeated sub-message
#   ("map", key_kind, value_kind)        → map field (value_kind may be
#                                          ("msg", schema))
# Decoded messages are plain dicts; absent fields take proto3 defaults.


def encode(schema: dict[str, tuple], msg: dict[str, Any]) -> bytes:
    out = bytearray()
    for name, spec in schema.items():
        num, kind = spec[0], spec[1]
        value = msg.get(name)
        if value is None:
            continue
        if kind in ("int64", "bool", "enum"):
            iv = int(value)
            if iv == 0:
                continue  # proto3 default omitted
            _write_varint(out, _tag(num, _VARINT))
            _write_varint(out, iv)
        elif kind == "double":
            packed8 = struct.pack("<d", float(value))
            if packed8 == b"\x00" * 8:
                continue  # proto3 default (+0.0) omitted; -0.0 is emitted
            _write_varint(out, _tag(num, _FIXED64))
            out += packed8
        elif kind == "string":
            data = value.encode("utf-8") if isinstance(value, str) else bytes(value)
            if not data:
                continue
            _write_varint(out, _tag(num, _LEN))
            _write_varint(out, len(data))
            out += data
        elif kind == "msg":
            sub = encode(spec[2], value)
            _write_varint(out, _tag(num, _LEN))
            _write_varint(out, len(sub))
            out += sub
        elif kind in ("rep_int64", "rep_enum"):
            if not value:
                continue
            packed = bytearray()
            for v in value:
                _write_varint(packed, int(v))
            _write_varint(out, _tag(num, _LEN))
            _write_varint(out, len(packed))
            out += packed
        elif kind == "rep_double":
            if not value:
                continue
            packed = struct.pack(f"<{len(value)}d", *[float(v) for v in value])
            _write_varint(out, _tag(num, _LEN))
            _write_varint(out, len(packed))
            out += packed
        elif kind == "rep_msg":
            for item in value:
                sub = encode(spec[2], item)
                _write_varint(out, _tag(num, _LEN))
                _write_varint(out, len(sub))
                out += sub
        elif kind == "map":
            key_kind, value_kind = spec[2], spec[3]
            entry_schema = {
                "key": (1, key_kind) if isinstance(key_kind, str) else (1,) + key_kind,
                "value": (2, value_kind)
                if isinstance(value_kind, str)
                else (2,) + value_kind,
            }
            for k, v in value.items():
                sub = encode(entry_schema, {"key": k, "value": v})
                _write_varint(out, _tag(num, _LEN))
                _write_varint(out, len(sub))
                out += sub
        else:
            raise TypeError(f"unknown kind {kind!r} for field {name}")
    return bytes(out)


def _default(kind: Any) -> Any:
    if kind in ("int64", "enum"):
        return 0
    if kind == "bool":
        return False
    if kind == "double":
        return 0.0
    if kind == "string":
        return ""
    if kind == "msg":
        return None
    if isinstance(kind, str) and kind.startswith("rep_"):
        return []
    return None


def decode(schema: dict[str, tuple], data: bytes) -> dict[str, Any]:
    by_num: dict[int, tuple[str, tuple]] = {spec[0]: (name, spec) for name, spec in schema.items()}
    msg: dict[str, Any] = {}
    for name, spec in schema.items():
        kind = spec[1]
        if kind == "map":
            msg[name] = {}
        elif kind in ("rep_int64", "rep_double", "rep_enum", "rep_msg"):
            msg[name] = []
        elif kind == "msg":
            msg[name] = None
        else:
            msg[name] = _default(kind)
    i = 0
    n = len(data)
    while i < n:
        tag, i = _read_varint(data, i)
        num, wt = tag >> 3, tag & 7
        entry = by_num.get(num)
        if entry is None:
            # skip unknown field
            if wt == _VARINT:
                _, i = _read_varint(data, i)
            elif wt == _FIXED64:
                i += 8
            elif wt == _LEN:
                ln, i = _read_varint(data, i)
                i += ln
            elif wt == 5:  # fixed32
                i += 4
            else:
                raise ValueError(f"unsupported wire type {wt}")
            continue
        name, spec = entry
        kind = spec[1]
        if wt == _VARINT:
            v, i = _read_varint(data, i)
            if kind == "bool":
                msg[name] = bool(v)
            elif kind in ("rep_int64", "rep_enum"):
                msg[name].append(v)
            else:
                msg[name] = v
        elif wt == _FIXED64:
            (v,) = struct.unpack_from("<d", data, i)
            i += 8
            if kind == "rep_double":
                msg[name].append(v)
            else:
                msg[name] = v
        elif wt == _LEN:
            ln, i = _read_varint(data, i)
            chunk = data[i : i + ln]
            i += ln
            if kind == "string":
                msg[name] = chunk.decode("utf-8")
            elif kind == "msg":
                msg[name] = decode(spec[2], chunk)
            elif kind == "rep_msg":
                msg[name].append(decode(spec[2], chunk))
            elif kind in ("rep_int64", "rep_enum"):
                j = 0
                while j < ln:
                    v, j = _read_varint(chunk, j)
                    msg[name].append(v)
            elif kind == "rep_double":
                msg[name].extend(
                    struct.unpack(f"<{ln // 8}d", chunk)
                )
            elif kind == "map":
                key_kind, value_kind = spec[2], spec[3]
                entry_schema = {
                    "key": (1, key_kind)
                    if isinstance(key_kind, str)
                    else (1,) + key_kind,
                    "value": (2, value_kind)
                    if isinstance(value_kind, str)
                    else (2,) + value_kind,
                }
                e = decode(entry_schema, chunk)
                msg[name][e["key"]] = e["value"]
            else:
                raise ValueError(f"length-delimited data for {name}:{kind}")
        else:
            raise ValueError(f"unsupported wire type {wt}")
    return msg


# ---- api.proto message schemas (field numbers from the reference proto) ------------

STUDY = {
    "study_id": (1, "int64"),
    "study_name": (2, "string"),
    "directions": (3, "rep_enum"),
    "user_attributes": (4, "map", "string", "string"),
    "system_attributes": (5, "map", "string", "string"),
}

TRIAL = {
    "trial_id": (1, "int64"),
    "number": (2, "int64"),
    "state": (3, "enum"),
    "values": (4, "rep_double"),
    "datetime_start": (5, "string"),
    "datetime_complete": (6, "string"),
    "params": (7, "map", "string", "double"),
    "distributions": (8, "map", "string", "string"),
    "user_attributes": (9, "map", "string", "string"),
    "system_attributes": (10, "map", "string", "string"),
    "intermediate_values": (11, "map", "int64", "double"),
}

# method name → (request schema, reply schema)
METHODS: dict[str, tuple[dict, dict]] = {
    "CreateNewStudy": (
        {"directions": (1, "rep_enum"), "study_name": (2, "string")},
        {"study_id": (1, "int64")},
    ),
    "DeleteStudy": ({"study_id": (1, "int64")}, {}),
    "SetStudyUserAttribute": (
        {"study_id": (1, "int64"), "key": (2, "string"), "value": (3, "string")},
        {},
    ),
    "SetStudySystemAttribute": (
        {"study_id": (1, "int64"), "key": (2, "string"), "value": (3, "string")},
        {},
    ),
    "GetStudyIdFromName": (
        {"study_name": (1, "string")},
        {"study_id": (1, "int64")},
    ),
    "GetStudyNameFromId": (
        {"study_id": (1, "int64")},
        {"study_name": (1, "string")},
    ),
    "GetStudyDirections": (
        {"study_id": (1, "int64")},
        {"directions": (1, "rep_enum")},
    ),
    "GetStudyUserAttributes": (
        {"study_id": (1, "int64")},
        {"user_attributes": (1, "map", "string", "string")},
    ),
    "GetStudySystemAttributes": (
        {"study_id": (1, "int64")},
        {"system_attributes": (1, "map", "string", "string")},
    ),
    "GetAllStudies": ({}, {"studies": (1, "rep_msg", STUDY)}),
    "CreateNewTrial": (
        {
            "study_id": (1, "int64"),
            "template_trial": (2, "msg", TRIAL),
            "template_trial_is_none": (3, "bool"),
        },
        {"trial_id": (1, "int64")},
    ),
    "SetTrialParameter": (
        {
            "trial_id": (1, "int64"),
            "param_name": (2, "string"),
            "param_value_internal": (3, "double"),
            "distribution": (4, "string"),
        },
        {},
    ),
    "GetTrialIdFromStudyIdTrialNumber": (
        {"study_id": (1, "int64"), "trial_number": (2, "int64")},
        {"trial_id": (1, "int64")},
    ),
    "SetTrialStateValues": (
        {"trial_id": (1, "int64"), "state": (2, "enum"), "values": (3, "rep_double")},
        {"trial_updated": (1, "bool")},
    ),
    "SetTrialIntermediateValue": (
        {
            "trial_id": (1, "int64"),
            "step": (2, "int64"),
            "intermediate_value": (3, "double"),
        },
        {},
    ),
    "SetTrialUserAttribute": (
        {"trial_id": (1, "int64"), "key": (2, "string"), "value": (3, "string")},
        {},
    ),
    "SetTrialSystemAttribute": (
        {"trial_id": (1, "int64"), "key": (2, "string"), "value": (3, "string")},
        {},
    ),
    "GetTrial": ({"trial_id": (1, "int64")}, {"trial": (1, "msg", TRIAL)}),
    "GetTrials": (
        {
            "study_id": (1, "int64"),
            "included_trial_ids": (2, "rep_int64"),
            "trial_id_greater_than": (3, "int64"),
        },
        {"trials": (1, "rep_msg", TRIAL)},
    ),
}

SERVICE = "optuna.StorageService"
