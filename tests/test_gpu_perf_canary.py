"""Perf canary (GPU): generous floors that only trip when the native HIP
path stops doing the work (e.g. a silent fallback or a broken kernel
launch).  Not a benchmark — bench.py is; these floors are ~2x below the
measured round-1 numbers to absorb box-to-box variance."""
import ctypes
import time

import pytest

import minio_amd
import oracle

pytestmark = pytest.mark.gpu


def test_encode_throughput_canary():
    d, p, bs, n = 8, 4, 1 << 20, 256
    lib = minio_amd._lib
    vp = ctypes.c_void_p
    with minio_amd.Erasure(d, p, bs) as e:
        ctx = e._ctx
        stride = lib.mec_shard_stride(bs, d)

        def check(st):
            assert st == 0, lib.mec_last_error()

        dev_data, dev_par, dev_sum = vp(), vp(), vp()
        check(lib.mec_dev_alloc(ctx, n * d * stride, ctypes.byref(dev_data)))
        check(lib.mec_dev_alloc(ctx, n * p * stride, ctypes.byref(dev_par)))
        check(lib.mec_dev_alloc(ctx, n * (d + p) * 32, ctypes.byref(dev_sum)))
        data = oracle.fill_random(min(n * bs, 64 << 20), 3)
        check(lib.mec_memcpy_h2d(ctx, dev_data, data, len(data)))
        # warm + timed
        for _ in range(2):
            check(lib.mec_encode_batch_dev(ctx, n, dev_data, bs, dev_par,
                                           minio_amd.HIGHWAYHASH256S, dev_sum))
        t0 = time.perf_counter()
        for _ in range(5):
            check(lib.mec_encode_batch_dev(ctx, n, dev_data, bs, dev_par,
                                           minio_amd.HIGHWAYHASH256S, dev_sum))
        dt = (time.perf_counter() - t0) / 5
        gibs = n * bs / dt / (1 << 30)
        # round-1 measures ~1000+ GiB/s at this batch; 250 = clearly broken
        assert gibs > 250, f"encode canary: {gibs:.0f} GiB/s"


def test_reconstruct_throughput_canary():
    d, p, bs, n = 8, 4, 1 << 20, 256
    lib = minio_amd._lib
    vp = ctypes.c_void_p
    with minio_amd.Erasure(d, p, bs) as e:
        ctx = e._ctx
        stride = lib.mec_shard_stride(bs, d)
        S = e.shard_size()

        def check(st):
            assert st == 0, lib.mec_last_error()

        dev_sh = vp()
        check(lib.mec_dev_alloc(ctx, n * (d + p) * stride,
                                ctypes.byref(dev_sh)))
        check(lib.mec_memset_dev(ctx, dev_sh, 7, n * (d + p) * stride))
        present = bytes([0] * 3 + [1] * (d + p - 3))
        for _ in range(2):
            check(lib.mec_reconstruct_batch_dev(ctx, n, dev_sh, present, S, 1))
        t0 = time.perf_counter()
        for _ in range(5):
            check(lib.mec_reconstruct_batch_dev(ctx, n, dev_sh, present, S, 1))
        dt = (time.perf_counter() - t0) / 5
        gibs = n * bs / dt / (1 << 30)
        assert gibs > 300, f"reconstruct canary: {gibs:.0f} GiB/s"
