"""Size-independent properties of the oracle, mirroring the reference's
behavioural tests (cmd/erasure_test.go:45-108 round trips,
cmd/erasure-heal_test.go erase patterns, cmd/bitrot_test.go shapes)."""
import itertools
import random

import pytest

import oracle


@pytest.mark.parametrize("d,p", [(2, 2), (4, 2), (8, 4), (12, 4), (16, 4)])
def test_roundtrip_erase_reconstruct(d, p):
    rng = random.Random(0x6D696E696F + d * 100 + p)
    data = bytes(rng.getrandbits(8) for _ in range(4096 + 13))
    rs = oracle.RS(d, p)
    shards = rs.encode_data(data)
    # every erase pattern of size p over a few random choices
    idxs = list(range(d + p))
    for _ in range(10):
        erased = rng.sample(idxs, p)
        damaged = [None if i in erased else s for i, s in enumerate(shards)]
        rec = rs.reconstruct(damaged, data_only=False)
        assert rec == shards, f"erased={erased}"


def test_split_padding_semantics():
    # Split: shard k = bytes [k*ceil(n/d), ...), tail zero-padded
    # (cmd/erasure-coding.go:81 -> reedsolomon Split)
    rs = oracle.RS(3, 2)
    data = bytes((i + 1) & 0xFF for i in range(256))  # 256 B, d=3 -> per=86, pad=2
    shards = rs.encode_data(data)
    assert len(shards[0]) == 86
    assert shards[0] == data[:86]
    assert shards[1] == data[86:172]
    assert shards[2] == data[172:] + b"\0\0"


def test_reconstruct_too_few():
    rs = oracle.RS(4, 2)
    shards = rs.encode_data(bytes(64))
    damaged = [None, None, None] + shards[3:]
    with pytest.raises(ValueError):
        rs.reconstruct(damaged)


def test_data_only_leaves_parity_missing():
    rs = oracle.RS(4, 2)
    shards = rs.encode_data(bytes(range(64)))
    damaged = [None] + shards[1:5] + [None]
    rec = rs.reconstruct(damaged, data_only=True)
    assert rec[0] == shards[0]
    assert rec[5] is None


def test_hh256_ragged_tails_vs_full():
    # ragged lengths exercise UpdateRemainder; shape check only (the exact
    # values are pinned transitively -- see oracle.h header note)
    for n in [0, 1, 3, 4, 15, 16, 17, 31, 32, 33, 63, 64, 100]:
        msg = bytes((i * 7 + 1) & 0xFF for i in range(n))
        s1 = oracle.bitrot_sum(oracle.HIGHWAYHASH256S, msg)
        s2 = oracle.bitrot_sum(oracle.HIGHWAYHASH256S, msg)
        assert s1 == s2 and len(s1) == 32
        if n > 0:
            s3 = oracle.bitrot_sum(oracle.HIGHWAYHASH256S, msg[:-1] + bytes([msg[-1] ^ 1]))
            assert s3 != s1


def test_shard_file_size_math():
    # cmd/erasure-coding.go:121-132 and cmd/bitrot.go:156-161
    bs, d = 1 << 20, 8
    S = oracle.ceil_frac(bs, d)
    assert S == 131072
    # 2.5 blocks
    total = 2 * bs + 12345
    want = 2 * S + oracle.ceil_frac(12345, d)
    got = (total // bs) * S + oracle.ceil_frac(total % bs, d)
    assert got == want
    assert oracle.bitrot_shard_file_size(35, 10, oracle.HIGHWAYHASH256S) == 4 * 32 + 35
    assert oracle.bitrot_shard_file_size(35, 10, oracle.SHA256) == 35


def test_encode_stream_layout():
    # [hash||shard]* per drive (cmd/bitrot-streaming.go:57-75)
    d, p, bs = 4, 2, 256
    data = oracle.fill_random(1000, 42)
    streams, _ = oracle.encode_stream(d, p, bs, data, oracle.HIGHWAYHASH256S)
    S = oracle.ceil_frac(bs, d)  # 64
    n_blocks = 4  # ceil(1000/256); last block 232 bytes -> S_last = 58
    rs = oracle.RS(d, p)
    for i, st in enumerate(streams):
        off = 0
        for b in range(n_blocks):
            blk = data[b * bs:(b + 1) * bs]
            shards = rs.encode_data(blk)
            h = st[off:off + 32]
            sh = st[off + 32:off + 32 + len(shards[i])]
            assert sh == shards[i]
            assert h == oracle.bitrot_sum(oracle.HIGHWAYHASH256S, sh)
            off += 32 + len(shards[i])
        assert off == len(st)


def test_simd_matches_scalar():
    """The SIMD bench legs (oracle/simd.c; BENCH ONLY, never the checker)
    must be bit-identical to the scalar restatement: GF constant-multiply
    for every coefficient, and the AVX2 HighwayHash main loop across
    lengths covering every remainder residue."""
    import ctypes
    isa = oracle.cpu_isa()
    assert isa in ("gfni+avx2", "avx2", "scalar")
    # GF: all 256 coefficients over a buffer covering all byte values,
    # with a ragged tail exercising the scalar cleanup
    n = 256 * 3 + 7
    data = bytes((i * 37 + 11) & 0xFF for i in range(n))
    acc0 = bytes((i * 101 + 5) & 0xFF for i in range(n))
    for c in range(256):
        want = bytes(a ^ oracle._lib.mo_gf_mul(c, b)
                     for a, b in zip(acc0, data)) if c else acc0
        got = oracle.gal_mul_xor_fast(c, data, acc0)
        assert got == want, f"coef {c} (isa={isa})"
    # HH: magic key, lengths covering 0..63 plus larger odd sizes
    key = bytes.fromhex(
        "4be734fa8e238acd263e83e6bb96855204 0f935da39f441497e09d1322de36a0"
        .replace(" ", ""))
    for ln in list(range(0, 64)) + [100, 1024, 4096 + 31, 87382]:
        msg = oracle.fill_random(ln, 0xABCDEF + ln)
        assert oracle.hh256_fast(key, msg) == oracle.bitrot_sum(
            oracle.HIGHWAYHASH256S, msg), f"len {ln} (isa={isa})"


def test_simd_reconstruct_matches_scalar():
    """mo_rs_reconstruct_fast (bench leg) == scalar checker on random
    erasure patterns."""
    import ctypes
    import random
    lib = oracle._lib
    lib.mo_rs_reconstruct_fast.argtypes = [
        ctypes.POINTER(oracle._MoRS), ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8)),
        ctypes.c_char_p, ctypes.c_size_t, ctypes.c_int]
    lib.mo_rs_reconstruct_fast.restype = ctypes.c_int
    rng = random.Random(5)
    for (d, p) in [(8, 4), (12, 4), (4, 2)]:
        bs = d * 997
        data = oracle.fill_random(bs, d * 7 + p)
        rs = oracle.RS(d, p)
        want = rs.encode_data(data)
        S = len(want[0])
        for _ in range(6):
            n_er = rng.randint(1, p)
            erased = rng.sample(range(d + p), n_er)
            bufs = [ctypes.create_string_buffer(
                        b"" if i in erased else want[i], S)
                    for i in range(d + p)]
            arr = (ctypes.POINTER(ctypes.c_uint8) * (d + p))(
                *[ctypes.cast(b, ctypes.POINTER(ctypes.c_uint8))
                  for b in bufs])
            present = bytes(0 if i in erased else 1 for i in range(d + p))
            rc = lib.mo_rs_reconstruct_fast(ctypes.byref(rs._rs), arr,
                                            present, S, 0)
            assert rc == 0
            for i in range(d + p):
                assert bufs[i].raw[:S] == want[i], (d, p, erased, i)


def test_encode_linearity_full_size():
    """GF(2^8) encode is linear over XOR: encode(x^y) == encode(x) ^
    encode(y), checked at the FULL headline shard size (131072 B) — a
    size-independent property that pins full-size behavior beyond the
    small golden vectors (contract: properties at BASELINE sizes)."""
    d, p = 8, 4
    bs = 1 << 20
    rs = oracle.RS(d, p)
    x = oracle.fill_random(bs, 0xAAAA)
    y = oracle.fill_random(bs, 0xBBBB)
    xy = bytes(a ^ b for a, b in zip(x, y))
    ex, ey, exy = (rs.encode_data(v) for v in (x, y, xy))
    for s in range(d + p):
        want = bytes(a ^ b for a, b in zip(ex[s], ey[s]))
        assert exy[s] == want, s
    # systematic prefix: data shards are the split input verbatim
    S = len(ex[0])
    for k in range(d):
        assert ex[k] == x[k * S:(k + 1) * S].ljust(S, b"\0"), k
