"""Reference implementation of the framework's LZ4-style byte codec.

The reference project compresses every wire payload with ``lz4.frame``
on top of ZFP (/root/reference/src/dispatcher.py:81-84, node.py:107).
This module is the CPU/bit-exact spec for the gfx950 HIP kernels in
``defer_amd/csrc/lz4.hip``: GPU compress output must equal
``compress(x)`` byte-for-byte (same greedy parse, same hash policy),
and ``decompress`` must invert both.

Format (all integers little-endian):
    header: u32 raw_len | u32 nblocks | u32 off[nblocks + 1]
    body:   per-4096-byte-block LZ4 sequence streams, block i at off[i]
Blocks compress independently (no cross-block matches) so the GPU can
assign one wavefront per block. Sequences use the standard LZ4 block
format: token byte (hi nibble = literal count, lo nibble = match length
minus 4, 15 = continue with 255-run extension bytes), literals, u16
little-endian backward offset, match-length extension bytes. The final
sequence of each block is literals-only; the decoder stops when the
block's raw size has been produced.
"""

import struct

import numpy as np

BLK = 4096
HBITS = 11
HSIZE = 1 << HBITS
_MULT = 2654435761


def _hash(v: int) -> int:
    return ((v * _MULT) & 0xFFFFFFFF) >> (32 - HBITS)


def _compress_block(buf: bytes) -> bytes:
    n = len(buf)
    out = bytearray()
    tab = [0xFFFF] * HSIZE
    pos = 0
    anchor = 0
    mend = n - 5          # matches may not extend into the last 5 bytes

    def read32(p):
        return struct.unpack_from("<I", buf, p)[0]

    def emit(lit_from, lit_len, moff, mlen):
        mtok = 0 if mlen == 0 else mlen - 4
        out.append((min(lit_len, 15) << 4) | min(mtok, 15))
        if lit_len >= 15:
            rem = lit_len - 15
            while True:
                out.append(min(rem, 255))
                if rem < 255:
                    break
                rem -= 255
        out.extend(buf[lit_from:lit_from + lit_len])
        if mlen == 0:
            return
        out.append(moff & 0xFF)
        out.append(moff >> 8)
        if mtok >= 15:
            rem = mtok - 15
            while True:
                out.append(min(rem, 255))
                if rem < 255:
                    break
                rem -= 255

    # LZ4-style skip acceleration: after every 64 failed probes the scan
    # step grows by one, bounding worst-case work on incompressible data.
    cnt = 64
    while pos < n - 8:
        v = read32(pos)
        h = _hash(v)
        cand = tab[h]
        tab[h] = pos
        if cand != 0xFFFF and read32(cand) == v:
            mlen = 4
            while pos + mlen < mend and buf[cand + mlen] == buf[pos + mlen]:
                mlen += 1
            emit(anchor, pos - anchor, pos - cand, mlen)
            pos += mlen
            anchor = pos
            cnt = 64
        else:
            pos += cnt >> 6
            cnt += 1
    emit(anchor, n - anchor, 0, 0)    # final literals-only sequence
    return bytes(out)


def _decompress_block(src: bytes, blen: int) -> bytes:
    dst = bytearray()
    p = 0
    while len(dst) < blen and p < len(src):
        tok = src[p]
        p += 1
        lit = tok >> 4
        if lit == 15:
            while True:
                e = src[p]
                p += 1
                lit += e
                if e != 255:
                    break
        dst += src[p:p + lit]
        p += lit
        if len(dst) >= blen:
            break
        moff = src[p] | (src[p + 1] << 8)
        p += 2
        mlen = (tok & 0xF) + 4
        if (tok & 0xF) == 15:
            while True:
                e = src[p]
                p += 1
                mlen += e
                if e != 255:
                    break
        for _ in range(mlen):          # byte-wise: overlap-correct LZ77
            dst.append(dst[-moff])
    return bytes(dst)


def compress(data) -> np.ndarray:
    """data: bytes-like or uint8 ndarray -> uint8 ndarray (full stream)."""
    if isinstance(data, np.ndarray):
        data = data.astype(np.uint8, copy=False).tobytes()
    n = len(data)
    nblocks = (n + BLK - 1) // BLK
    blocks = [_compress_block(data[i * BLK:min(n, (i + 1) * BLK)])
              for i in range(nblocks)]
    offs = [0]
    for b in blocks:
        offs.append(offs[-1] + len(b))
    hdr = struct.pack(f"<II{nblocks + 1}I", n, nblocks, *offs)
    return np.frombuffer(hdr + b"".join(blocks), dtype=np.uint8)


def decompress(comp) -> np.ndarray:
    if isinstance(comp, np.ndarray):
        comp = comp.astype(np.uint8, copy=False).tobytes()
    n, nblocks = struct.unpack_from("<II", comp, 0)
    offs = struct.unpack_from(f"<{nblocks + 1}I", comp, 8)
    body = 4 * (2 + nblocks + 1)
    out = bytearray()
    for i in range(nblocks):
        blen = min(BLK, n - i * BLK)
        src = comp[body + offs[i]:body + offs[i + 1]]
        out += _decompress_block(src, blen)
    return np.frombuffer(bytes(out), dtype=np.uint8)
