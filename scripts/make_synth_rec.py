"""Pack a synthetic ImageNet-shaped RecordIO file (raw records) for the
data-pipeline benchmark: bench.py --rec <out.rec>."""
import argparse
import struct

import numpy as np

MAGIC = 0xced7230a


def main():
    p = argparse.ArgumentParser()
    p.add_argument('out')
    p.add_argument('--n', type=int, default=512)
    p.add_argument('--hw', type=int, default=256)
    p.add_argument('--seed', type=int, default=0)
    a = p.parse_args()
    rs = np.random.RandomState(a.seed)
    with open(a.out, 'wb') as f:
        for i in range(a.n):
            img = rs.randint(0, 256, (a.hw, a.hw, 3), dtype=np.uint8)
            payload = struct.pack('<II', a.hw, a.hw) + img.tobytes()
            body = struct.pack('<IfQQ', 0, float(i % 1000), 0, 0) + payload
            f.write(struct.pack('<II', MAGIC, len(body)))
            f.write(body)
            f.write(b'\x00' * ((-len(body)) % 4))
    print(f'wrote {a.n} raw records to {a.out}')


if __name__ == '__main__':
    main()
