"""Three-way BLAKE3 cross-check.

BLAKE3 is the least-pinned primitive (two official vectors from memory +
dual C/C++ implementations).  This adds a THIRD independent implementation,
written here in pure Python directly from the BLAKE3 paper's compression
definition, and requires all three to agree on random inputs across the
single-chunk size range this path uses (plus XOF prefixes)."""
import random

IV = [0x6A09E667, 0xBB67AE85, 0x3C6EF372, 0xA54FF53A,
      0x510E527F, 0x9B05688C, 0x1F83D9AB, 0x5BE0CD19]
MSG_PERM = [2, 6, 3, 10, 7, 0, 4, 13, 1, 11, 12, 5, 9, 14, 15, 8]
CHUNK_START, CHUNK_END, ROOT = 1, 2, 8
M32 = 0xFFFFFFFF


def _rotr(x, n):
    return ((x >> n) | (x << (32 - n))) & M32


def _g(v, a, b, c, d, mx, my):
    v[a] = (v[a] + v[b] + mx) & M32
    v[d] = _rotr(v[d] ^ v[a], 16)
    v[c] = (v[c] + v[d]) & M32
    v[b] = _rotr(v[b] ^ v[c], 12)
    v[a] = (v[a] + v[b] + my) & M32
    v[d] = _rotr(v[d] ^ v[a], 8)
    v[c] = (v[c] + v[d]) & M32
    v[b] = _rotr(v[b] ^ v[c], 7)


def _compress(h, m, t, blen, flags):
    v = h[:8] + IV[:4] + [t & M32, (t >> 32) & M32, blen, flags]
    m = list(m)
    for r in range(7):
        _g(v, 0, 4, 8, 12, m[0], m[1])
        _g(v, 1, 5, 9, 13, m[2], m[3])
        _g(v, 2, 6, 10, 14, m[4], m[5])
        _g(v, 3, 7, 11, 15, m[6], m[7])
        _g(v, 0, 5, 10, 15, m[8], m[9])
        _g(v, 1, 6, 11, 12, m[10], m[11])
        _g(v, 2, 7, 8, 13, m[12], m[13])
        _g(v, 3, 4, 9, 14, m[14], m[15])
        if r < 6:
            m = [m[MSG_PERM[i]] for i in range(16)]
    return [v[i] ^ v[i + 8] for i in range(8)] + \
           [v[i + 8] ^ h[i] for i in range(8)]


def blake3_py(msg: bytes, outlen: int = 32) -> bytes:
    assert len(msg) <= 1024, "single-chunk reference"
    h = IV[:]
    nblocks = max(1, (len(msg) + 63) // 64)
    for b in range(nblocks - 1):
        blk = msg[b * 64:(b + 1) * 64]
        m = [int.from_bytes(blk[4 * i:4 * i + 4], "little")
             for i in range(16)]
        h = _compress(h, m, 0, 64, CHUNK_START if b == 0 else 0)[:8]
    last = msg[(nblocks - 1) * 64:]
    blen = len(last)
    last = last.ljust(64, b"\0")
    m = [int.from_bytes(last[4 * i:4 * i + 4], "little") for i in range(16)]
    flags = (CHUNK_START if nblocks == 1 else 0) | CHUNK_END | ROOT
    out = b""
    t = 0
    while len(out) < outlen:
        words = _compress(h, m, t, blen, flags)
        out += b"".join(w.to_bytes(4, "little") for w in words)
        t += 1
    return out[:outlen]


def test_three_way_agreement(oracle):
    import gsm_amd
    eng = gsm_amd.Engine()
    rng = random.Random(0xB3)
    lengths = [0, 1, 31, 32, 33, 44, 49, 63, 64, 65, 127, 128, 129, 512,
               1023, 1024]
    for n in lengths:
        msg = bytes(rng.randrange(256) for _ in range(n))
        py = blake3_py(msg)
        assert oracle.blake3(msg) == py, f"oracle != py at len {n}"
        assert eng.selftest_blake3(msg) == py, f"engine != py at len {n}"


def test_xof_three_way(oracle):
    msg = b"xof-cross-check"
    py = blake3_py(msg, 200)
    assert oracle.blake3(msg, outlen=200) == py


def test_official_vectors_python_impl():
    assert blake3_py(b"").hex() == (
        "af1349b9f5f9a1a6a0404dea36dcc9499bcb25c9adc112b7cc9a93cae41f3262")
    assert blake3_py(bytes([0])).hex() == (
        "2d3adedff11b61f14c886e35afa036736dcd87a74d27b5c1510225d0f592e213")
