# Debug probe (r2): reproduces and localizes fused3 parity mismatches —
# prints (block, parity-row, first/last bad offset, count) per trial.
# Used to find the launcher/kernel G mismatch.
import os, sys
sys.path.insert(0, os.environ.get("R", "/root/repo"))
import minio_amd, oracle
SEED = 0x6D696E696F
d, p, bs = 8, 4, 64 * 8 * 1024
n = 32
with minio_amd.Erasure(d, p, bs) as e:
    blocks = [oracle.fill_random(bs, SEED + 200 + b) for b in range(n)]
    for trial in range(3):
        shards, sums = e.encode_batch(b"".join(blocks), bs, n,
                                      minio_amd.HIGHWAYHASH256S)
        ors = oracle.RS(d, p)
        bad = []
        for b in range(n):
            osh = ors.encode_data(blocks[b])
            for i in range(p):
                got, want = shards[b][d+i], osh[d+i]
                if got != want:
                    offs = [o for o in range(len(want)) if got[o] != want[o]]
                    bad.append((b, i, offs[0], offs[-1], len(offs)))
            for s in range(d+p):
                w = oracle.bitrot_sum(oracle.HIGHWAYHASH256S, osh[s])
                if sums[b][s] != w:
                    bad.append(("sum", b, s))
        print(f"trial {trial}: n_bad={len(bad)}")
        for x in bad[:12]: print("   ", x)
