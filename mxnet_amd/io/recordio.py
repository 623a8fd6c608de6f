"""RecordIO file format (reference python/mxnet/recordio.py + dmlc-core
recordio).  Byte format: per record

    u32 magic 0xced7230a
    u32 lrecord: upper 3 bits = cflag, lower 29 bits = length
    payload bytes, padded to 4-byte boundary

Multi-part records (cflag 1/2/3) supported on read.  IRHeader packs
(flag, label, id, id2) ahead of image payloads (pack/unpack).
"""
import os
import struct

import numpy as _np

_MAGIC = 0xced7230a
_LEN_MASK = (1 << 29) - 1


class MXRecordIO:
    def __init__(self, uri, flag):
        self.uri = uri
        self.flag = flag
        self.open()

    def open(self):
        self.fp = open(self.uri, 'wb' if self.flag == 'w' else 'rb')

    def close(self):
        self.fp.close()

    def reset(self):
        self.fp.seek(0)

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    def write(self, buf):
        assert self.flag == 'w'
        self.fp.write(struct.pack('<II', _MAGIC, len(buf)))
        self.fp.write(buf)
        pad = (-len(buf)) % 4
        if pad:
            self.fp.write(b'\x00' * pad)

    def read(self):
        assert self.flag == 'r'
        header = self.fp.read(8)
        if len(header) < 8:
            return None
        magic, lrec = struct.unpack('<II', header)
        if magic != _MAGIC:
            raise IOError(f'{self.uri}: bad record magic {magic:#x}')
        cflag = lrec >> 29
        length = lrec & _LEN_MASK
        buf = self.fp.read(length)
        pad = (-length) % 4
        if pad:
            self.fp.read(pad)
        if cflag in (0,):
            return buf
        # multi-part record: keep reading continuation parts
        parts = [buf]
        while cflag in (1, 2):
            magic, lrec = struct.unpack('<II', self.fp.read(8))
            cflag = lrec >> 29
            length = lrec & _LEN_MASK
            parts.append(self.fp.read(length))
            pad = (-length) % 4
            if pad:
                self.fp.read(pad)
        return b''.join(parts)

    def tell(self):
        return self.fp.tell()


class MXIndexedRecordIO(MXRecordIO):
    def __init__(self, idx_path, uri, flag):
        self.idx_path = idx_path
        self.idx = {}
        self.keys = []
        super().__init__(uri, flag)
        if flag == 'r' and os.path.exists(idx_path):
            with open(idx_path) as f:
                for line in f:
                    key, pos = line.strip().split('\t')
                    self.idx[int(key)] = int(pos)
                    self.keys.append(int(key))

    def close(self):
        if self.flag == 'w':
            with open(self.idx_path, 'w') as f:
                for k in self.keys:
                    f.write(f'{k}\t{self.idx[k]}\n')
        super().close()

    def write_idx(self, idx, buf):
        pos = self.tell()
        self.write(buf)
        self.idx[idx] = pos
        self.keys.append(idx)

    def read_idx(self, idx):
        self.fp.seek(self.idx[idx])
        return self.read()


# IRHeader: (flag u32, label f32, id u64, id2 u64)
_IR_FORMAT = '<IfQQ'
_IR_SIZE = struct.calcsize(_IR_FORMAT)


class IRHeader:
    def __init__(self, flag, label, id, id2):
        self.flag, self.label, self.id, self.id2 = flag, label, id, id2


def pack(header, s):
    label = header.label
    if isinstance(label, (list, tuple, _np.ndarray)):
        arr = _np.asarray(label, dtype=_np.float32)
        hdr = struct.pack(_IR_FORMAT, len(arr), 0.0, header.id, header.id2)
        return hdr + arr.tobytes() + s
    return struct.pack(_IR_FORMAT, 0, float(label), header.id, header.id2) + s


def unpack(s):
    flag, label, id_, id2 = struct.unpack(_IR_FORMAT, s[:_IR_SIZE])
    payload = s[_IR_SIZE:]
    if flag > 0:
        arr = _np.frombuffer(payload[:4 * flag], dtype=_np.float32)
        return IRHeader(flag, arr, id_, id2), payload[4 * flag:]
    return IRHeader(flag, label, id_, id2), payload
