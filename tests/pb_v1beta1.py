"""Hand-rolled protobuf wire helpers for the kubelet DevicePlugin v1beta1
messages — an independent Python mirror of native/grpc/proto.h +
native/deviceplugin/dp_messages.h used to drive the C++ plugin from grpcio
with identity (de)serializers."""

from __future__ import annotations


def put_varint(out: bytearray, v: int) -> None:
    while v >= 0x80:
        out.append((v & 0x7F) | 0x80)
        v >>= 7
    out.append(v)


def put_tag(out: bytearray, field: int, wire: int) -> None:
    put_varint(out, (field << 3) | wire)


def put_bytes(out: bytearray, field: int, payload: bytes) -> None:
    put_tag(out, field, 2)
    put_varint(out, len(payload))
    out.extend(payload)


def put_str(out: bytearray, field: int, s: str) -> None:
    put_bytes(out, field, s.encode())


def read_varint(buf: bytes, pos: int):
    v = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        v |= (b & 0x7F) << shift
        if not b & 0x80:
            return v, pos
        shift += 7


def fields(buf: bytes):
    """Yield (field, wire, value) where value is int (wire 0) or bytes (2)."""
    pos = 0
    while pos < len(buf):
        tag, pos = read_varint(buf, pos)
        field, wire = tag >> 3, tag & 7
        if wire == 0:
            v, pos = read_varint(buf, pos)
            yield field, wire, v
        elif wire == 2:
            ln, pos = read_varint(buf, pos)
            yield field, wire, buf[pos:pos + ln]
            pos += ln
        elif wire == 5:
            yield field, wire, buf[pos:pos + 4]
            pos += 4
        elif wire == 1:
            yield field, wire, buf[pos:pos + 8]
            pos += 8
        else:
            raise ValueError(f"bad wire type {wire}")


# ---- encoders for requests the fake kubelet / test client sends ----

def encode_allocate_request(per_container_ids):
    out = bytearray()
    for ids in per_container_ids:
        cr = bytearray()
        for i in ids:
            put_str(cr, 1, i)
        put_bytes(out, 1, bytes(cr))
    return bytes(out)


def encode_preferred_request(available, must_include, size):
    cr = bytearray()
    for i in available:
        put_str(cr, 1, i)
    for i in must_include:
        put_str(cr, 2, i)
    put_tag(cr, 3, 0)
    put_varint(cr, size)
    out = bytearray()
    put_bytes(out, 1, bytes(cr))
    return bytes(out)


# ---- decoders for responses / the Register request ----

def decode_register_request(buf: bytes) -> dict:
    out = {"version": "", "endpoint": "", "resource_name": "", "options": {}}
    for f, w, v in fields(buf):
        if f == 1:
            out["version"] = v.decode()
        elif f == 2:
            out["endpoint"] = v.decode()
        elif f == 3:
            out["resource_name"] = v.decode()
        elif f == 4:
            for f2, _, v2 in fields(v):
                if f2 == 1:
                    out["options"]["pre_start_required"] = bool(v2)
                if f2 == 2:
                    out["options"]["get_preferred_allocation_available"] = bool(v2)
    return out


def decode_options(buf: bytes) -> dict:
    out = {"pre_start_required": False,
           "get_preferred_allocation_available": False}
    for f, w, v in fields(buf):
        if f == 1:
            out["pre_start_required"] = bool(v)
        if f == 2:
            out["get_preferred_allocation_available"] = bool(v)
    return out


def decode_device(buf: bytes) -> dict:
    d = {"id": "", "health": "", "numa": None}
    for f, w, v in fields(buf):
        if f == 1:
            d["id"] = v.decode()
        elif f == 2:
            d["health"] = v.decode()
        elif f == 3:
            for f2, _, v2 in fields(v):
                if f2 == 1:
                    for f3, _, v3 in fields(v2):
                        if f3 == 1:
                            d["numa"] = v3
    return d


def decode_list_and_watch(buf: bytes) -> list:
    return [decode_device(v) for f, w, v in fields(buf) if f == 1]


def decode_map_entry(buf: bytes):
    k = v = b""
    for f, _, val in fields(buf):
        if f == 1:
            k = val
        if f == 2:
            v = val
    return k.decode(), v.decode()


def decode_allocate_response(buf: bytes) -> list:
    containers = []
    for f, w, v in fields(buf):
        if f != 1:
            continue
        cr = {"envs": {}, "mounts": [], "devices": [], "annotations": {}}
        for f2, _, v2 in fields(v):
            if f2 == 1:
                k, val = decode_map_entry(v2)
                cr["envs"][k] = val
            elif f2 == 2:
                m = {}
                for f3, _, v3 in fields(v2):
                    if f3 == 1:
                        m["container_path"] = v3.decode()
                    if f3 == 2:
                        m["host_path"] = v3.decode()
                    if f3 == 3:
                        m["read_only"] = bool(v3)
                cr["mounts"].append(m)
            elif f2 == 3:
                d = {}
                for f3, _, v3 in fields(v2):
                    if f3 == 1:
                        d["container_path"] = v3.decode()
                    if f3 == 2:
                        d["host_path"] = v3.decode()
                    if f3 == 3:
                        d["permissions"] = v3.decode()
                cr["devices"].append(d)
            elif f2 == 4:
                k, val = decode_map_entry(v2)
                cr["annotations"][k] = val
        containers.append(cr)
    return containers


def decode_preferred_response(buf: bytes) -> list:
    out = []
    for f, w, v in fields(buf):
        if f == 1:
            out.append([v2.decode() for f2, _, v2 in fields(v) if f2 == 1])
    return out
