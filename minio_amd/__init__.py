"""minio_amd — MI355X-native implementation of MinIO's erasure-coding +
bitrot hot path.

The product boundary is the C-ABI in include/minio_ec.h, implemented by
libminio_ec_hip.so (HIP/gfx950 kernels + C++ host drivers).  This package is
a thin ctypes veneer over that boundary, mirroring the reference's Go
surfaces 1:1 for tests and benchmarks:

  - Erasure             <-> cmd/erasure-coding.go:35-141 (NewErasure,
                            EncodeData, DecodeDataBlocks,
                            DecodeDataAndParityBlocks, ShardSize,
                            ShardFileSize, ShardFileOffset)
  - Erasure.encode_stream / decode_stream / heal_stream
                        <-> Erasure.Encode / Decode / Heal
                            (cmd/erasure-encode.go:76, erasure-decode.go:239,
                             :317) over the streaming bitrot on-disk layout
                            (cmd/bitrot-streaming.go)
  - bitrot_verify       <-> bitrotVerify (cmd/bitrot.go:164-216)

No CPU fallback exists: constructing an Erasure on a machine without a
visible GPU raises NoGPUError (MEC_ERR_NO_GPU).  The host-side mirrors of
the shard-size math are pure integer functions and work anywhere.
"""
import ctypes
import os

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_HERE, "libminio_ec_hip.so")

# Bitrot algorithm ids (cmd/xl-storage-format-v1.go:146-153)
SHA256 = 1
HIGHWAYHASH256 = 2
HIGHWAYHASH256S = 3
BLAKE2B512 = 4
DEFAULT_BITROT_ALGORITHM = HIGHWAYHASH256S

_HASH_SIZE = {SHA256: 32, HIGHWAYHASH256: 32, HIGHWAYHASH256S: 32, BLAKE2B512: 64}

# mec_status values (include/minio_ec.h)
MEC_OK = 0
_STATUS_NAMES = {
    1: "ErrInvShardNum",
    2: "ErrMaxShardNum",
    3: "ErrTooFewShards",
    4: "ErrShortData",
    5: "errFileCorrupt",
    6: "InvalidArgument",
    7: "HIPError",
    8: "NoGPU",
}


class MecError(RuntimeError):
    def __init__(self, status, detail=""):
        self.status = status
        name = _STATUS_NAMES.get(status, str(status))
        super().__init__(f"minio_ec: {name}{(': ' + detail) if detail else ''}")


class FileCorruptError(MecError):
    pass


class NoGPUError(MecError):
    pass


def _load():
    if not os.path.exists(_LIB_PATH):
        raise ImportError(
            f"{_LIB_PATH} not built — run `make -C minio_amd/csrc` "
            "(or __graft_entry__.build()).  The product path requires the "
            "HIP extension; there is no fallback."
        )
    lib = ctypes.CDLL(_LIB_PATH)
    c = ctypes
    i64, i32, sz = c.c_int64, c.c_int, c.c_size_t
    u8p, vp = c.c_char_p, c.c_void_p
    lib.mec_last_error.restype = c.c_char_p
    lib.mec_shard_size.restype = i64
    lib.mec_shard_size.argtypes = [i64, i32]
    lib.mec_shard_file_size.restype = i64
    lib.mec_shard_file_size.argtypes = [i64, i32, i64]
    lib.mec_shard_file_offset.restype = i64
    lib.mec_shard_file_offset.argtypes = [i64, i32, i64, i64, i64]
    lib.mec_bitrot_shard_file_size.restype = i64
    lib.mec_bitrot_shard_file_size.argtypes = [i64, i64, i32]
    lib.mec_shard_stride.restype = i64
    lib.mec_shard_stride.argtypes = [i64, i32]
    lib.mec_ctx_create.argtypes = [i32, i32, i64, i32, c.POINTER(vp)]
    lib.mec_ctx_destroy.argtypes = [vp]
    lib.mec_encode_batch.argtypes = [vp, i32, u8p, i64, u8p, i32, u8p]
    lib.mec_encode_batch_dev.argtypes = [vp, i32, vp, i64, vp, i32, vp]
    lib.mec_encode_batch_dev_async.argtypes = [vp, i32, vp, i64, vp, i32, vp]
    lib.mec_encode_batch_dev_pipe.argtypes = [vp, i32, vp, i64, vp, i32, vp]
    lib.mec_pipe_sync.argtypes = [vp]
    lib.mec_reconstruct_batch.argtypes = [vp, i32, u8p, u8p, i64, i32]
    lib.mec_reconstruct_batch_dev.argtypes = [vp, i32, vp, u8p, i64, i32]
    lib.mec_reconstruct_batch_dev_async.argtypes = [vp, i32, vp, u8p, i64, i32]
    lib.mec_bitrot_sum_batch.argtypes = [vp, i32, i32, u8p, i64, i64, u8p]
    lib.mec_bitrot_sum_batch_dev.argtypes = [vp, i32, i32, vp, i64, i64, vp]
    lib.mec_bitrot_verify_batch.argtypes = [vp, i32, i32, u8p, i64, i64, u8p, u8p]
    lib.mec_bitrot_verify_stream.argtypes = [vp, u8p, i64, i64, i32, u8p, i64]
    lib.mec_encode_stream.argtypes = [vp, u8p, i64, i32, c.POINTER(u8p), u8p]
    lib.mec_decode_stream.argtypes = [vp, c.POINTER(u8p), u8p, i32, i64, i64, i64, u8p]
    lib.mec_heal_stream.argtypes = [vp, c.POINTER(u8p), i32, i64, c.POINTER(u8p)]
    lib.mec_dev_alloc.argtypes = [vp, sz, c.POINTER(vp)]
    lib.mec_dev_free.argtypes = [vp, vp]
    lib.mec_memcpy_h2d.argtypes = [vp, vp, u8p, sz]
    lib.mec_memcpy_d2h.argtypes = [vp, u8p, vp, sz]
    lib.mec_memset_dev.argtypes = [vp, vp, i32, sz]
    lib.mec_stream_sync.argtypes = [vp]
    lib.mec_timer_start.argtypes = [vp]
    lib.mec_timer_stop.argtypes = [vp, c.POINTER(c.c_float)]
    return lib


_lib = _load()


def _check(status):
    if status == MEC_OK:
        return
    detail = (_lib.mec_last_error() or b"").decode()
    if status == 5:
        raise FileCorruptError(status, detail)
    if status == 8:
        raise NoGPUError(status, detail)
    raise MecError(status, detail)


def device_count() -> int:
    return _lib.mec_device_count()


def hash_size(algo: int) -> int:
    return _HASH_SIZE[algo]


# ---- host-side shard-size math (pure, no GPU) -----------------------------

def shard_size(block_size: int, d: int) -> int:
    return _lib.mec_shard_size(block_size, d)


def shard_file_size(block_size: int, d: int, total_length: int) -> int:
    return _lib.mec_shard_file_size(block_size, d, total_length)


def shard_file_offset(block_size: int, d: int, start: int, length: int,
                      total: int) -> int:
    return _lib.mec_shard_file_offset(block_size, d, start, length, total)


def bitrot_shard_file_size(size: int, shard_size_: int, algo: int) -> int:
    return _lib.mec_bitrot_shard_file_size(size, shard_size_, algo)


class Erasure:
    """Mirror of the reference Erasure type (cmd/erasure-coding.go:35).

    Requires a visible GPU (raises NoGPUError otherwise)."""

    def __init__(self, data_blocks: int, parity_blocks: int,
                 block_size: int = 1 << 20, device: int = 0):
        self.d = data_blocks
        self.p = parity_blocks
        self.block_size = block_size
        self._ctx = ctypes.c_void_p()
        _check(_lib.mec_ctx_create(data_blocks, parity_blocks, block_size,
                                   device, ctypes.byref(self._ctx)))

    def close(self):
        if self._ctx:
            _lib.mec_ctx_destroy(self._ctx)
            self._ctx = ctypes.c_void_p()

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()

    def _ck(self):
        if not self._ctx:
            raise MecError(6, "context is closed")
        return self._ctx

    # -- size math (instance mirrors) --
    def shard_size(self) -> int:
        return shard_size(self.block_size, self.d)

    def shard_file_size(self, total_length: int) -> int:
        return shard_file_size(self.block_size, self.d, total_length)

    def shard_file_offset(self, start: int, length: int, total: int) -> int:
        return shard_file_offset(self.block_size, self.d, start, length, total)

    # -- EncodeData (cmd/erasure-coding.go:77-89) --
    def encode_data(self, data: bytes, algo: int = None):
        """Split + Encode one block.  Returns (shards, sums):
        shards = d+p bytes objects of ceil(len/d) (data shards zero-padded),
        sums = per-shard bitrot digests when algo given, else None."""
        if len(data) == 0:
            return [b""] * (self.d + self.p), None
        shards, sums = self.encode_batch(data, len(data), 1, algo)
        return ([sh for sh in shards[0]],
                sums[0] if sums is not None else None)

    def encode_batch(self, blocks: bytes, block_len: int, n: int,
                     algo: int = None):
        """n independent blocks packed at block_len stride.  Returns
        (shards, sums): shards[b][s] bytes, sums[b][s] digests or None."""
        S = shard_size(block_len, self.d)
        parity = ctypes.create_string_buffer(n * self.p * S)
        sums_buf = None
        if algo is not None:
            sums_buf = ctypes.create_string_buffer(
                n * (self.d + self.p) * _HASH_SIZE[algo])
        _check(_lib.mec_encode_batch(self._ck(), n, blocks, block_len, parity,
                                     algo or 0, sums_buf))
        out = []
        hs = _HASH_SIZE[algo] if algo is not None else 0
        sums = [] if algo is not None else None
        for b in range(n):
            blk = blocks[b * block_len:(b + 1) * block_len]
            shards = []
            for k in range(self.d):
                sh = blk[k * S:(k + 1) * S]
                shards.append(sh + b"\0" * (S - len(sh)))
            for i in range(self.p):
                off = (b * self.p + i) * S
                shards.append(parity.raw[off:off + S])
            out.append(shards)
            if algo is not None:
                t = self.d + self.p
                sums.append([
                    sums_buf.raw[(b * t + s) * hs:(b * t + s + 1) * hs]
                    for s in range(t)
                ])
        return out, sums

    # -- DecodeDataBlocks / DecodeDataAndParityBlocks --
    def decode_data_blocks(self, shards):
        """cmd/erasure-coding.go:94-107: None entries = missing; returns the
        list with data shards filled.  No-op when nothing is missing."""
        return self._reconstruct(shards, data_only=True)

    def decode_data_and_parity_blocks(self, shards):
        """cmd/erasure-coding.go:111-113."""
        return self._reconstruct(shards, data_only=False)

    def _reconstruct(self, shards, data_only):
        total = self.d + self.p
        assert len(shards) == total
        missing = [i for i, s in enumerate(shards) if s is None or len(s) == 0]
        if not missing or len(missing) == total:
            return list(shards)
        per = next(len(s) for s in shards if s)
        buf = ctypes.create_string_buffer(total * per)
        present = bytes(0 if (s is None or len(s) == 0) else 1 for s in shards)
        for i, s in enumerate(shards):
            if s:
                buf[i * per:(i + 1) * per] = s
        _check(_lib.mec_reconstruct_batch(self._ck(), 1, buf, present, per,
                                          1 if data_only else 0))
        out = []
        for i in range(total):
            if present[i] or not data_only or i < self.d:
                out.append(buf.raw[i * per:(i + 1) * per])
            else:
                out.append(shards[i])
        return out

    # -- streaming-format drivers (Erasure.Encode / Decode / Heal) --
    def encode_stream(self, src: bytes, algo: int = HIGHWAYHASH256S):
        """Returns (drive_streams, whole_sums): d+p per-drive on-disk streams
        ([hash||shard]* for HighwayHash256S), whole-file digests for the
        legacy algorithms."""
        total = self.d + self.p
        fsz = bitrot_shard_file_size(
            self.shard_file_size(len(src)), self.shard_size(), algo)
        bufs = [ctypes.create_string_buffer(max(fsz, 1)) for _ in range(total)]
        arr = (ctypes.c_char_p * total)(*[
            ctypes.cast(b, ctypes.c_char_p) for b in bufs])
        ws = None
        if algo != HIGHWAYHASH256S:
            ws = ctypes.create_string_buffer(total * _HASH_SIZE[algo])
        _check(_lib.mec_encode_stream(self._ck(), src, len(src), algo, arr, ws))
        streams = [b.raw[:fsz] for b in bufs]
        sums = None
        if ws is not None:
            hs = _HASH_SIZE[algo]
            sums = [ws.raw[i * hs:(i + 1) * hs] for i in range(total)]
        return streams, sums

    def decode_stream(self, drive_streams, total_length: int, offset: int,
                      length: int, algo: int = HIGHWAYHASH256S,
                      whole_sums=None):
        """drive_streams: list of d+p streams (None = missing drive)."""
        total = self.d + self.p
        arr = (ctypes.c_char_p * total)(*[
            s if s is not None else None for s in drive_streams])
        ws = None
        if whole_sums is not None:
            ws = b"".join(s if s else b"\0" * _HASH_SIZE[algo]
                          for s in whole_sums)
        dst = ctypes.create_string_buffer(max(length, 1))
        _check(_lib.mec_decode_stream(self._ck(), arr, ws, algo, total_length,
                                      offset, length, dst))
        return dst.raw[:length]

    def heal_stream(self, drive_streams, total_length: int,
                    algo: int = HIGHWAYHASH256S):
        """Regenerate missing drives' streams (Erasure.Heal).  Returns the
        full list with reconstructed entries for drives that were None."""
        total = self.d + self.p
        fsz = bitrot_shard_file_size(
            self.shard_file_size(total_length), self.shard_size(), algo)
        arr = (ctypes.c_char_p * total)(*[
            s if s is not None else None for s in drive_streams])
        outs = []
        outp = (ctypes.c_char_p * total)()
        for i, s in enumerate(drive_streams):
            if s is None:
                b = ctypes.create_string_buffer(max(fsz, 1))
                outs.append(b)
                outp[i] = ctypes.cast(b, ctypes.c_char_p)
            else:
                outs.append(None)
                outp[i] = None
        _check(_lib.mec_heal_stream(self._ck(), arr, algo, total_length, outp))
        return [
            drive_streams[i] if drive_streams[i] is not None
            else outs[i].raw[:fsz]
            for i in range(total)
        ]

    # -- batch hashing --
    def bitrot_sum_batch(self, algo: int, msgs: bytes, msg_len: int,
                         msg_stride: int, n: int):
        hs = _HASH_SIZE[algo]
        out = ctypes.create_string_buffer(max(n * hs, 1))
        _check(_lib.mec_bitrot_sum_batch(self._ck(), algo, n, msgs, msg_len,
                                         msg_stride, out))
        return [out.raw[i * hs:(i + 1) * hs] for i in range(n)]

    def bitrot_verify_stream(self, stream: bytes, part_size: int, algo: int,
                             want_sum: bytes = None,
                             shard_size_: int = None) -> bool:
        """bitrotVerify (cmd/bitrot.go:164-216).  True = intact."""
        ss = shard_size_ if shard_size_ is not None else self.shard_size()
        try:
            _check(_lib.mec_bitrot_verify_stream(
                self._ck(), stream, len(stream), part_size, algo, want_sum, ss))
            return True
        except FileCorruptError:
            return False
