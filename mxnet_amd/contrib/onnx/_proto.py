"""Minimal protobuf wire-format codec for ONNX ModelProto.

The image has no ``onnx`` package, so the ONNX serialization is done
directly at the protobuf wire level (the field numbers below are the
stable public ``onnx.proto`` schema).  Only the subset the exporter /
importer uses is implemented: varint + length-delimited fields, and the
message types Model/Graph/Node/Attribute/Tensor/ValueInfo.

Reference parity: python/mxnet/contrib/onnx (mx2onnx/onnx2mx) produced
onnx ModelProtos through the onnx python package; this module replaces
that dependency with a self-contained codec.
"""
import struct

# ---------------------------------------------------------------------------
# wire primitives
# ---------------------------------------------------------------------------


def _varint(n):
    if n < 0:
        n += 1 << 64  # two's-complement 64-bit
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _tag(field, wt):
    return _varint((field << 3) | wt)


def vint(field, v):
    return _tag(field, 0) + _varint(int(v))


def blob(field, payload):
    return _tag(field, 2) + _varint(len(payload)) + payload


def string(field, s):
    return blob(field, s.encode('utf-8'))


def flt(field, v):
    return _tag(field, 5) + struct.pack('<f', float(v))


def packed_floats(field, vals):
    return blob(field, b''.join(struct.pack('<f', float(v)) for v in vals))


def packed_ints(field, vals):
    return blob(field, b''.join(_varint(int(v)) for v in vals))


# ---------------------------------------------------------------------------
# decoding
# ---------------------------------------------------------------------------


def _read_varint(buf, pos):
    result = shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7


def parse(buf):
    """Parse a message into {field_no: [raw values]} (varint -> int,
    len-delimited -> bytes, fixed32/64 -> bytes)."""
    fields = {}
    pos = 0
    while pos < len(buf):
        key, pos = _read_varint(buf, pos)
        field, wt = key >> 3, key & 7
        if wt == 0:
            v, pos = _read_varint(buf, pos)
        elif wt == 2:
            ln, pos = _read_varint(buf, pos)
            v = buf[pos:pos + ln]
            pos += ln
        elif wt == 5:
            v = buf[pos:pos + 4]
            pos += 4
        elif wt == 1:
            v = buf[pos:pos + 8]
            pos += 8
        else:
            raise ValueError(f'unsupported wire type {wt}')
        fields.setdefault(field, []).append(v)
    return fields


def as_int(fields, n, default=0):
    v = fields.get(n)
    return int(v[0]) if v else default


def as_sint(fields, n, default=0):
    v = as_int(fields, n, default)
    return v - (1 << 64) if v >= (1 << 63) else v


def as_str(fields, n, default=''):
    v = fields.get(n)
    return v[0].decode('utf-8') if v else default


def as_bytes(fields, n, default=b''):
    v = fields.get(n)
    return v[0] if v else default


def as_float(fields, n, default=0.0):
    v = fields.get(n)
    return struct.unpack('<f', v[0])[0] if v else default


def unpack_ints(raw):
    """Packed repeated varint payload -> list of ints."""
    out, pos = [], 0
    while pos < len(raw):
        v, pos = _read_varint(raw, pos)
        out.append(v - (1 << 64) if v >= (1 << 63) else v)
    return out


def repeated_ints(fields, n):
    """Repeated int64 field: packed or unpacked encodings both occur."""
    out = []
    for v in fields.get(n, []):
        if isinstance(v, (bytes, bytearray)):
            out.extend(unpack_ints(v))
        else:
            out.append(v - (1 << 64) if v >= (1 << 63) else v)
    return out


def repeated_floats(fields, n):
    out = []
    for v in fields.get(n, []):
        if isinstance(v, (bytes, bytearray)) and len(v) != 4:
            out.extend(struct.unpack(f'<{len(v)//4}f', v))
        elif isinstance(v, (bytes, bytearray)):
            out.append(struct.unpack('<f', v)[0])
    return out


# ---------------------------------------------------------------------------
# ONNX field numbers (public onnx.proto schema)
# ---------------------------------------------------------------------------

# TensorProto.DataType
FLOAT, UINT8, INT8, INT32, INT64 = 1, 2, 3, 6, 7
BOOL, FLOAT16, DOUBLE, BFLOAT16 = 9, 10, 11, 16

# AttributeProto.AttributeType
A_FLOAT, A_INT, A_STRING, A_TENSOR = 1, 2, 3, 4
A_GRAPH, A_FLOATS, A_INTS, A_STRINGS = 5, 6, 7, 8


def tensor_proto(name, dims, data_type, raw):
    return (packed_ints(1, dims) + vint(2, data_type) + string(8, name) +
            blob(9, raw))


def attr_float(name, v):
    return string(1, name) + flt(2, v) + vint(20, A_FLOAT)


def attr_int(name, v):
    return string(1, name) + vint(3, v) + vint(20, A_INT)


def attr_str(name, s):
    return string(1, name) + blob(4, s.encode('utf-8')) + vint(20, A_STRING)


def attr_ints(name, vals):
    return string(1, name) + packed_ints(8, vals) + vint(20, A_INTS)


def node_proto(op_type, inputs, outputs, name, attrs=b''):
    body = b''.join(string(1, i) for i in inputs)
    body += b''.join(string(2, o) for o in outputs)
    body += string(3, name) + string(4, op_type)
    if attrs:
        body += attrs  # pre-encoded repeated AttributeProto field 5
    return body


def attr_field(encoded_attr):
    return blob(5, encoded_attr)


def value_info(name, elem_type, shape):
    dims = b''.join(blob(1, vint(1, d)) for d in shape)
    ttype = vint(1, elem_type) + blob(2, dims)
    return string(1, name) + blob(2, blob(1, ttype))


def graph_proto(nodes, name, initializers, inputs, outputs):
    body = b''.join(blob(1, n) for n in nodes)
    body += string(2, name)
    body += b''.join(blob(5, t) for t in initializers)
    body += b''.join(blob(11, vi) for vi in inputs)
    body += b''.join(blob(12, vi) for vi in outputs)
    return body


def model_proto(graph, opset=17, producer='mxnet_amd'):
    return (vint(1, 8) +                      # ir_version 8
            string(2, producer) +
            blob(8, vint(2, opset)) +         # opset_import {version}
            blob(7, graph))
