# ORACLE — TEST INFRASTRUCTURE ONLY (see oracle.h).
# ctypes wrapper over liboracle.so.  Only tests/, __graft_entry__.smoke()
# and bench.py's cpu_baseline leg may import this package.  It is the
# parity checker for the product HIP path, never the product path itself.
import ctypes
import os

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_HERE, "liboracle.so")
if not os.path.exists(_LIB_PATH):
    raise ImportError(
        "oracle/liboracle.so not built — run `make -C oracle` (or __graft_entry__.build())"
    )
_lib = ctypes.CDLL(_LIB_PATH)

_P8 = ctypes.POINTER(ctypes.c_uint8)


class _MoRS(ctypes.Structure):
    _fields_ = [
        ("d", ctypes.c_int),
        ("p", ctypes.c_int),
        ("matrix", ctypes.c_uint8 * (256 * 256)),
    ]


_lib.mo_rs_init.argtypes = [ctypes.POINTER(_MoRS), ctypes.c_int, ctypes.c_int]
_lib.mo_rs_encode.argtypes = [ctypes.POINTER(_MoRS), ctypes.POINTER(_P8), ctypes.c_size_t]
_lib.mo_rs_reconstruct.argtypes = [
    ctypes.POINTER(_MoRS), ctypes.POINTER(_P8), ctypes.c_char_p,
    ctypes.c_size_t, ctypes.c_int,
]
_lib.mo_rs_reconstruct.restype = ctypes.c_int
_lib.mo_bitrot_sum.argtypes = [ctypes.c_int, ctypes.c_char_p, ctypes.c_size_t, ctypes.c_char_p]
_lib.mo_bitrot_size.argtypes = [ctypes.c_int]
_lib.mo_bitrot_size.restype = ctypes.c_int
_lib.mo_xxh64.restype = ctypes.c_uint64
_lib.mo_xxh64.argtypes = [ctypes.c_char_p, ctypes.c_size_t, ctypes.c_uint64]
_lib.mo_fill_random.argtypes = [ctypes.c_char_p, ctypes.c_size_t, ctypes.c_uint64]
_lib.mo_cpu_encode_bench.restype = ctypes.c_double
_lib.mo_cpu_encode_bench.argtypes = [
    ctypes.c_int, ctypes.c_int, ctypes.c_size_t, ctypes.c_int,
    ctypes.c_int, ctypes.c_int, ctypes.c_uint64,
]
_lib.mo_cpu_isa.restype = ctypes.c_char_p
_lib.mo_cpu_isa.argtypes = []
_lib.mo_gal_mul_xor_fast.argtypes = [
    ctypes.c_uint8, ctypes.c_char_p, ctypes.c_char_p, ctypes.c_size_t,
]
_lib.mo_hh256_fast.argtypes = [
    ctypes.c_char_p, ctypes.c_char_p, ctypes.c_size_t, ctypes.c_char_p,
]
_lib.mo_gf_mul.restype = ctypes.c_uint8
_lib.mo_gf_mul.argtypes = [ctypes.c_uint8, ctypes.c_uint8]
_lib.mo_cpu_reconstruct_bench.restype = ctypes.c_double
_lib.mo_cpu_reconstruct_bench.argtypes = [
    ctypes.c_int, ctypes.c_int, ctypes.c_size_t, ctypes.c_int,
    ctypes.c_int, ctypes.c_int, ctypes.c_uint64,
]

SHA256 = 1
HIGHWAYHASH256 = 2
HIGHWAYHASH256S = 3
BLAKE2B512 = 4


def ceil_frac(n: int, d: int) -> int:
    """cmd/utils.go:689 for positive operands."""
    return -(-n // d) if d else 0


class RS:
    """Reed-Solomon oracle for one (d, p) geometry.

    Restates reedsolomon.Encoder as used via Erasure
    (reference cmd/erasure-coding.go:61-113)."""

    def __init__(self, d: int, p: int):
        self._rs = _MoRS()
        if _lib.mo_rs_init(ctypes.byref(self._rs), d, p) != 0:
            raise ValueError(f"invalid geometry d={d} p={p}")
        self.d, self.p = d, p

    @property
    def matrix(self):
        """(d+p) x d encode matrix rows as list of bytes."""
        rows = []
        for r in range(self.d + self.p):
            rows.append(bytes(self._rs.matrix[r * self.d:(r + 1) * self.d]))
        return rows

    def split(self, data: bytes):
        """reedsolomon.Encoder.Split padding semantics
        (cmd/erasure-coding.go:81): d shards of ceil(len/d), tail zero-padded.
        Returns list of d bytes objects."""
        if len(data) == 0:
            raise ValueError("short data")
        per = ceil_frac(len(data), self.d)
        padded = data + b"\0" * (per * self.d - len(data))
        return [padded[k * per:(k + 1) * per] for k in range(self.d)]

    def encode_blocks(self, data_shards):
        """Parity shards for the given d data shards (Encoder.Encode)."""
        per = len(data_shards[0])
        bufs = [(ctypes.c_uint8 * per).from_buffer_copy(s) for s in data_shards]
        bufs += [(ctypes.c_uint8 * per)() for _ in range(self.p)]
        arr = (_P8 * (self.d + self.p))(*[ctypes.cast(b, _P8) for b in bufs])
        _lib.mo_rs_encode(ctypes.byref(self._rs), arr, per)
        return [bytes(b) for b in bufs[self.d:]]

    def encode_data(self, data: bytes):
        """Erasure.EncodeData (cmd/erasure-coding.go:77-89): Split + Encode.
        Returns all d+p shards."""
        ds = self.split(data)
        return ds + self.encode_blocks(ds)

    def reconstruct(self, shards, data_only=False):
        """shards: list of d+p entries, None marks missing.  Returns the
        complete list (ReconstructData when data_only, else Reconstruct;
        cmd/erasure-coding.go:94-113)."""
        total = self.d + self.p
        assert len(shards) == total
        per = next(len(s) for s in shards if s is not None)
        present = bytes(1 if s is not None else 0 for s in shards)
        bufs = [
            (ctypes.c_uint8 * per).from_buffer_copy(s) if s is not None
            else (ctypes.c_uint8 * per)()
            for s in shards
        ]
        arr = (_P8 * total)(*[ctypes.cast(b, _P8) for b in bufs])
        rc = _lib.mo_rs_reconstruct(ctypes.byref(self._rs), arr, present, per,
                                    1 if data_only else 0)
        if rc == -2:
            raise ValueError("too few shards")
        if rc != 0:
            raise RuntimeError(f"reconstruct failed rc={rc}")
        out = [bytes(b) for b in bufs]
        if data_only:
            # parity slots that were missing stay missing (ReconstructData)
            for i in range(self.d, total):
                if shards[i] is None:
                    out[i] = None
        return out


def bitrot_sum(algo: int, msg: bytes) -> bytes:
    size = _lib.mo_bitrot_size(algo)
    out = ctypes.create_string_buffer(size)
    _lib.mo_bitrot_sum(algo, msg, len(msg), out)
    return out.raw


def bitrot_size(algo: int) -> int:
    return _lib.mo_bitrot_size(algo)


def xxh64(msg: bytes, seed: int = 0) -> int:
    return _lib.mo_xxh64(msg, len(msg), seed)


def fill_random(n: int, seed: int) -> bytes:
    buf = ctypes.create_string_buffer(n)
    _lib.mo_fill_random(buf, n, seed)
    return buf.raw


def bitrot_shard_file_size(size: int, shard_size: int, algo: int) -> int:
    """cmd/bitrot.go:156-161."""
    if algo != HIGHWAYHASH256S:
        return size
    return ceil_frac(size, shard_size) * bitrot_size(algo) + size


def encode_stream(d: int, p: int, block_size: int, data: bytes, algo: int):
    """Full per-drive streaming-format oracle: Erasure.Encode block loop
    (cmd/erasure-encode.go:76-108) + streamingBitrotWriter.Write
    (cmd/bitrot-streaming.go:44-75).  Returns d+p per-drive byte streams in
    the on-disk [hash||shard]* layout (hash only for HighwayHash256S;
    whole-file algorithms return (streams, whole_sums))."""
    rs = RS(d, p)
    streams = [b""] * (d + p)
    whole = [[] for _ in range(d + p)] if algo != HIGHWAYHASH256S else None
    for off in range(0, max(len(data), 1), block_size):
        block = data[off:off + block_size]
        if not block and off > 0:
            break
        shards = rs.encode_data(block) if block else [b""] * (d + p)
        for i, s in enumerate(shards):
            if algo == HIGHWAYHASH256S:
                streams[i] += bitrot_sum(algo, s) + s
            else:
                streams[i] += s
                whole[i].append(s)
    if algo == HIGHWAYHASH256S:
        return streams, None
    return streams, [bitrot_sum(algo, b"".join(w)) for w in whole]


def cpu_isa() -> str:
    """ISA path the SIMD bench legs dispatch to (simd.c): "gfni+avx2",
    "avx2", or "scalar"."""
    return _lib.mo_cpu_isa().decode()


def gal_mul_xor_fast(c: int, data: bytes, acc: bytes) -> bytes:
    """SIMD bench-leg GF op: returns acc ^= c*data (bytes)."""
    out = ctypes.create_string_buffer(acc, len(acc))
    _lib.mo_gal_mul_xor_fast(c, data, out, len(data))
    return out.raw[:len(data)]


def hh256_fast(key32: bytes, msg: bytes) -> bytes:
    out = ctypes.create_string_buffer(32)
    _lib.mo_hh256_fast(key32, msg, len(msg), out)
    return out.raw
