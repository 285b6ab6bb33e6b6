"""Oracle vs the reference's own boot-time golden vectors.

These pin the CPU oracle bit-exactly to the reference:
 - erasureSelfTest fingerprints (cmd/erasure-coding.go:149-206): xxh64 over
   index||shard of the full encode of bytes 0..255, 60 (d,p) configs, plus
   the delete-shard-0-and-reconstruct check.
 - bitrotSelfTest chained digests (cmd/bitrot.go:224-255) for
   SHA256 / BLAKE2b512 / HighwayHash256(S) with the magic key.
"""
import json
import os

HERE = os.path.dirname(os.path.abspath(__file__))

import oracle


def test_erasure_selftest_fingerprints():
    golden = json.load(open(os.path.join(HERE, "golden/erasure_selftest.json")))
    data = bytes(range(256))
    for key, want in golden["fingerprints"].items():
        d, p = map(int, key.split(","))
        rs = oracle.RS(d, p)
        shards = rs.encode_data(data)
        stream = b"".join(bytes([i]) + s for i, s in enumerate(shards))
        got = f"{oracle.xxh64(stream):016x}"
        assert got == want, f"d={d} p={p}"
        # reconstruct check (cmd/erasure-coding.go:192-200)
        first = shards[0]
        shards2 = [None] + shards[1:]
        rec = rs.reconstruct(shards2, data_only=True)
        assert rec[0] == first, f"reconstruct d={d} p={p}"


def test_bitrot_selftest_digests():
    golden = json.load(open(os.path.join(HERE, "golden/bitrot_selftest.json")))
    block_sizes = {"SHA256": (oracle.SHA256, 64),
                   "HighwayHash256": (oracle.HIGHWAYHASH256, 32),
                   "HighwayHash256S": (oracle.HIGHWAYHASH256S, 32),
                   "BLAKE2b512": (oracle.BLAKE2B512, 128)}
    for name, (algo, block) in block_sizes.items():
        size = oracle.bitrot_size(algo)
        msg = b""
        sum_ = b""
        for _ in range(block):
            sum_ = oracle.bitrot_sum(algo, msg)
            msg += sum_
        assert sum_.hex() == golden["digests"][name], name
