#!/usr/bin/env python3
"""bench.py — measures BASELINE.json's metric: EC encode+decode GiB/s,
RS(k=8, m=3), 1 MiB chunks, on 1..8 MI355X.

One step = one encode pass + one decode pass (3 erasures) of the hot path
over a device-resident 4096-stripe batch per GPU (configs[1] of
BASELINE.json; decode erasure count = m per configs[2] style). `value` is
whole-job input-byte throughput with the reference benchmark's accounting
(seconds per iterations * input bytes, ceph_erasure_code_benchmark.cc:193):
encode processes k*C*S input bytes, decode processes k*C*S, so one step
accounts 2*k*C*S per GPU. Inputs are device-resident random bytes
(seed 0xEC via splitmix64 fill) — BASELINE.md requires non-constant fill.

Multi-GPU: stripes shard; each rank owns its batch end-to-end (weak
scaling); torch.distributed (RCCL) is used only for barriers and the
max-over-ranks timing reduction — no data-path collective.

Every measured run is preceded by a bit-exact parity check of sampled
stripes against the CPU oracle (BASELINE.md "Parity").
"""
import argparse
import json
import os
import sys
import time

import numpy as np

ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, ROOT)

GIB = 1024 ** 3


def splitmix64(idx, seed):
    """numpy replica of the device fill (ec_core.hip ecx_splitmix64)."""
    x = (np.uint64(seed) ^ idx.astype(np.uint64)) + np.uint64(0x9E3779B97F4A7C15)
    with np.errstate(over="ignore"):
        x = (x ^ (x >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
        x = (x ^ (x >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
    return x ^ (x >> np.uint64(31))


def expected_fill(byte_off, nbytes, seed):
    """Bytes the device fill kernel wrote at [byte_off, byte_off+nbytes)."""
    assert byte_off % 8 == 0 and nbytes % 8 == 0
    idx = np.arange(byte_off // 8, (byte_off + nbytes) // 8, dtype=np.uint64)
    return splitmix64(idx, seed).view(np.uint8)


def parity_selfcheck(ctx, dptr, args, seed, sample_stripes=2):
    """Download a few encoded stripes, re-encode with the oracle from the
    deterministic fill, compare bit-exactly."""
    import oracle
    k, m, C, S = args.k, args.m, args.chunk_bytes, args.stripes
    n = k + m
    stripe_bytes = n * C
    rng = np.random.default_rng(123)
    for s in sorted(rng.choice(S, size=min(sample_stripes, S), replace=False)):
        host = np.zeros(stripe_bytes, dtype=np.uint8)
        # download stripe s
        import ctypes
        import ceph_amd
        src = ctypes.c_void_p(dptr.value + int(s) * stripe_bytes)
        ctx.download(host, src)
        data = [host[i * C:(i + 1) * C] for i in range(k)]
        # data region must equal the deterministic fill
        exp = expected_fill(int(s) * stripe_bytes, k * C, seed)
        if not np.array_equal(host[:k * C], exp):
            raise AssertionError(f"stripe {s}: data region != expected fill")
        if args.technique in ("cauchy_orig", "cauchy_good"):
            want = oracle.bitmatrix_encode(k, m, data, args.packetsize,
                                           technique=args.technique)
        elif args.technique == "jerasure_reed_sol_van_w16":
            want = oracle.encode_w16(k, m, data)
        else:
            want = oracle.encode(args.technique, k, m, data)
        for j in range(m):
            got = host[(k + j) * C:(k + j + 1) * C]
            if not np.array_equal(got, want[j]):
                raise AssertionError(
                    f"stripe {s}: GPU parity {j} mismatches oracle")


def _set_omp_threads(n):
    import ctypes
    try:
        ctypes.CDLL("libgomp.so.1").omp_set_num_threads(int(n))
        return True
    except OSError:
        return False


def cpu_baseline(args, budget_s=10.0):
    """Time the oracle's ISA-L-class AVX2/OpenMP path (kind='port') on a
    bounded sample of the same workload on this host's cores. Thread count
    is CALIBRATED: GPU-box hosts expose 256 hardware threads but collapse
    under full oversubscription (224 GiB/s at 64 threads vs 7 GiB/s at
    256, measured), so we pick the best of a few counts and report it."""
    import oracle
    k, m, C = args.k, args.m, args.chunk_bytes
    S = max(1, min(args.stripes, int(2 * GIB / ((k + m) * C))))  # <=2 GiB data
    batch = np.empty(S * (k + m) * C, dtype=np.uint8)
    oracle.cpu_first_touch(batch)  # NUMA-spread pages before content fill
    batch[:] = np.frombuffer(os.urandom(1 << 20), np.uint8).repeat(
        (batch.nbytes + (1 << 20) - 1) // (1 << 20))[:batch.nbytes]
    present = np.ones(k + m, np.uint8)
    present[sorted(np.random.default_rng(0xEC).choice(
        k + m, size=args.erasures, replace=False))] = 0

    hw = len(os.sched_getaffinity(0))
    candidates = sorted({min(hw, n) for n in (8, 16, 32, 64)})
    best_nt, best = candidates[0], 0.0
    calib = {}
    oracle.cpu_encode_batch(args.technique, k, m, batch, S, C)  # warm
    for nt in candidates:
        if not _set_omp_threads(nt):
            break
        # SUSTAINED sample per candidate (>=1.2 s of the combined loop):
        # burst iterations mislead — a GPU-box host measured 197 GiB/s in
        # sub-second bursts at 64 threads but 46 sustained (r1 verdict
        # item: the 58-vs-224 cpu_baseline gap), while 32 threads held
        # ~86-100. Sustained rates pick the honest count.
        t0 = time.perf_counter()
        iters = 0
        while time.perf_counter() - t0 < 1.2:
            oracle.cpu_encode_batch(args.technique, k, m, batch, S, C)
            oracle.cpu_decode_batch(args.technique, k, m, batch, present,
                                    S, C)
            iters += 1
        r = iters / (time.perf_counter() - t0)
        calib[nt] = round(r * 2 * k * C * S / GIB, 1)
        if r > best:
            best, best_nt = r, nt
    _set_omp_threads(best_nt)

    t0 = time.perf_counter()
    iters = 0
    while time.perf_counter() - t0 < budget_s:
        oracle.cpu_encode_batch(args.technique, k, m, batch, S, C)
        oracle.cpu_decode_batch(args.technique, k, m, batch, present, S, C)
        iters += 1
    dt = time.perf_counter() - t0
    gibs = iters * 2 * k * C * S / GIB / dt
    return {
        "value": round(gibs, 3),
        "unit": "GiB/s",
        "cores": best_nt,
        "kind": "port",
        "sample": (f"{iters}x encode+decode of a {S}-stripe batch "
                   f"(k={k},m={m},C={C}) in {dt:.1f}s; NUMA-spread pages; "
                   f"thread calibration {calib} GiB/s on a {hw}-thread "
                   "host"),
    }


def bench_mixed(args, dist=None, device=0):
    """BASELINE configs[4]: mixed k/m and 64 KiB..4 MiB chunks, streamed
    encode+decode with a per-(sub-batch) latency histogram. Shards across
    ranks like the main bench (weak scaling: every rank runs the full shape
    set on its own GPU). dist/device come from main() which handles the
    torch-before-ceph_amd runtime ordering."""
    import ceph_amd
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))

    # (k, m, chunk_bytes): BASELINE's k in {4,6,8,12}, m in {2,3,4},
    # chunks 64 KiB..4 MiB; sub-batch sized ~2 GiB of data each
    shapes = [(4, 2, 64 << 10), (6, 3, 256 << 10), (8, 3, 1 << 20),
              (12, 4, 4 << 20)]
    rng = np.random.default_rng(args.seed)
    ctxs = []
    for (k, m, C) in shapes:
        S = max(8, int(2 * GIB // (k * C)))
        ctx = ceph_amd.EcContext(k, m, "reed_sol_van", device=device,
                                 n_streams=args.streams)
        nbytes = S * (k + m) * C
        d = ctx.dbuf_alloc(nbytes)
        ctx.fill_random(d, nbytes, args.seed + rank)
        ctx.sync()
        n = k + m
        er = sorted(rng.choice(n, size=min(m, 3), replace=False).tolist())
        mask = (1 << n) - 1
        for e in er:
            mask &= ~(1 << e)
        ctxs.append((ctx, d, k, m, C, S, mask))

    def step(lat=None):
        for (ctx, d, k, m, C, S, mask) in ctxs:
            t0 = time.perf_counter()
            ctx.encode_batch(d, S, C)
            ctx.decode_batch(d, S, C, mask)
            ctx.sync()
            if lat is not None:
                lat.append((k, m, C, (time.perf_counter() - t0) * 1e3))

    for _ in range(args.warmup):
        step()
    if dist:
        import torch
        dist.barrier()
        torch.cuda.synchronize()
    lat = []
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step(lat)
    elapsed = time.perf_counter() - t0
    if dist:
        import torch
        torch.cuda.synchronize()
        t = torch.tensor([elapsed], dtype=torch.float64, device="cuda")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        bytes_per_step = sum(2 * k * C * S for (_, _, k, m, C, S, _) in ctxs)
        ms = sorted(x[3] for x in lat)
        hist = {
            "p10_ms": round(ms[int(len(ms) * .10)], 3),
            "p50_ms": round(ms[len(ms) // 2], 3),
            "p90_ms": round(ms[int(len(ms) * .90)], 3),
            "p99_ms": round(ms[min(len(ms) - 1, int(len(ms) * .99))], 3),
            "per_shape_ms": {f"k{k}m{m}c{C//1024}k": round(
                float(np.mean([x[3] for x in lat if x[:3] == (k, m, C)])), 3)
                for (k, m, C) in {x[:3] for x in lat}},
        }
        value = world * bytes_per_step * args.steps / GIB / elapsed
        print(json.dumps({
            "metric": "EC encode+decode GiB/s",
            "value": round(value, 2), "unit": "GiB/s", "n_gpus": world,
            "steps": args.steps, "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True, "scaling": "weak",
            "vs_baseline": None, "dtype": "u8", "data": "synthetic",
            "config": {"workload": "mixed k/m 64KiB-4MiB streamed "
                                   "encode+decode (BASELINE configs[4])",
                       "shapes": [f"k{k}m{m}c{C >> 10}KiB_s{S}"
                                  for (_, _, k, m, C, S, _) in ctxs],
                       "seed": hex(args.seed)},
            "latency_histogram": hist,
        }))
    for (ctx, d, *_rest) in ctxs:
        ctx.dbuf_free(d)
        ctx.close()
    if dist:
        dist.destroy_process_group()


def bench_slices(args):
    """Caller-shaped batching: the OSD write path makes many
    small variable-blocksize encode_chunks calls (shard_extent_map_t::
    encode, ECUtil.cc:485-514, EC_ALIGN=4096). Measures N device-resident
    OSD-sized slices encoded in ONE ecx_encode_slices launch vs one launch
    per slice — the number that says why batching matters on a GPU."""
    import ceph_amd
    k, m = args.k, args.m
    n = k + m
    rng = np.random.default_rng(args.seed)
    # OSD-shaped slice sizes: 4 KiB .. 64 KiB, EC_ALIGN-ish
    sizes = [int(rng.choice([4096, 8192, 16384, 32768, 65536]))
             for _ in range(4096)]
    total = sum(sz * n for sz in sizes)
    ctx = ceph_amd.EcContext(k, m, args.technique, device=0,
                             n_streams=args.streams)
    d = ctx.dbuf_alloc(total)
    ctx.fill_random(d, total, args.seed)
    ctx.sync()
    ptrs, off = [], 0
    for sz in sizes:
        for c in range(n):
            ptrs.append(d.value + off + c * sz)
        off += sz * n
    data_bytes = sum(sz * k for sz in sizes)

    def run_batched(steps):
        t0 = time.perf_counter()
        for _ in range(steps):
            ctx.encode_slices(ptrs, sizes)
            ctx.sync()
        return (time.perf_counter() - t0) / steps

    def run_per_slice(steps, limit=256):
        # one ecx_encode_slices launch per slice (the unbatched shape);
        # bounded subset — the point is per-launch overhead
        t0 = time.perf_counter()
        for _ in range(steps):
            o = 0
            for i, sz in enumerate(sizes[:limit]):
                ctx.encode_slices(ptrs[i * n:(i + 1) * n], [sz])
            ctx.sync()
        dt = (time.perf_counter() - t0) / steps
        frac = sum(sz * k for sz in sizes[:limit])
        return dt, frac

    run_batched(2)  # warm
    bt = run_batched(args.steps)
    pt, pbytes = run_per_slice(max(2, args.steps // 2))
    batched_gibs = data_bytes / GIB / bt
    unbatched_gibs = pbytes / GIB / pt
    print(json.dumps({
        "metric": "EC slice-batched encode GiB/s",
        "value": round(batched_gibs, 2), "unit": "GiB/s", "n_gpus": 1,
        "steps": args.steps, "warmup": 2,
        "ms_per_step": round(bt * 1e3, 3), "higher_is_better": True,
        "scaling": "weak", "vs_baseline": None, "dtype": "u8",
        "data": "synthetic",
        "config": {"workload": "4096 OSD-shaped slices (4-64 KiB) per "
                               "launch, device-resident (SURVEY a9)",
                   "k": k, "m": m, "n_slices": len(sizes),
                   "slices_per_second": round(len(sizes) / bt),
                   "unbatched_gibs": round(unbatched_gibs, 3),
                   "batched_vs_unbatched": round(
                       batched_gibs / unbatched_gibs, 1),
                   "seed": hex(args.seed)},
    }))
    ctx.dbuf_free(d)
    ctx.close()


def lrc_layers(k, m, l):
    """kml layer expansion (ErasureCodeLrc.cc:292-395): returns
    (chunk_count, [(data_positions, coding_positions), ...]) — global layer
    first, then local layers."""
    groups = (k + m) // l
    assert (k + m) % l == 0 and k % groups == 0 and m % groups == 0
    mapping = ""
    for _ in range(groups):
        mapping += "D" * (k // groups) + "_" * (m // groups) + "_"
    maps = []
    gmap = ""
    for _ in range(groups):
        gmap += "D" * (k // groups) + "c" * (m // groups) + "_"
    maps.append(gmap)
    for i in range(groups):
        lm = ""
        for j in range(groups):
            lm += ("D" * l + "c") if i == j else "_" * (l + 1)
        maps.append(lm)
    layers = []
    for smap in maps:
        data = [i for i, c in enumerate(smap) if c == "D"]
        coding = [i for i, c in enumerate(smap) if c == "c"]
        layers.append((data, coding))
    return len(mapping), layers


def bench_lrc(args):
    """BASELINE configs[3]-shaped LRC layered encode on device-resident
    batches: each layer is one ecx_matmul_batch over the layer's chunk ids
    with the product's reed_sol_van rows. BASELINE names k=8 m=3 l=4,
    which the reference's own parse_kml REJECTS ((k+m)%l != 0,
    ERROR_LRC_K_M_MODULO) — the nearest valid shape k=9 m=3 l=4 is used
    and noted."""
    import ceph_amd
    k, m, l = 9, 3, 4
    C = 512 * 1024  # configs[3]: 512 KiB chunks
    n, layers = lrc_layers(k, m, l)
    S = min(args.stripes, int(24 * GIB / (n * C)))
    # one ctx per layer shape for matrices; one ctx sized (n_total) for the
    # batch (k,m of the ctx only bound chunk ids: use k=n-? simplest: a ctx
    # with k+m == n)
    ctx = ceph_amd.EcContext(max(2, n - m), min(m, n - 2), "reed_sol_van",
                             device=0, n_streams=args.streams)
    nbytes = S * n * C
    d = ctx.dbuf_alloc(nbytes)
    ctx.fill_random(d, nbytes, args.seed)
    ctx.sync()
    # product-side layer matrices (reed_sol_van rows for (k_l, m_l))
    layer_rows = []
    for (data, coding) in layers:
        t = ceph_amd.EcContext(len(data), len(coding), "reed_sol_van",
                               device=0)
        g = t.matrix()
        layer_rows.append(np.ascontiguousarray(g[len(data):]))
        t.close()

    def step():
        for (data, coding), rows in zip(layers, layer_rows):
            ctx.matmul_batch(d, S, C, data, coding, rows)
        ctx.sync()

    step()  # warm + correctness pass below
    if not args.no_selfcheck:
        import oracle
        host = np.zeros(n * C, dtype=np.uint8)
        ctx.download(host, d)  # stripe 0
        chunks = [host[i * C:(i + 1) * C].copy() for i in range(n)]
        for (data, coding) in layers:
            want = oracle.encode_with_rows(
                oracle.matrix("reed_sol_van", len(data),
                              len(coding))[len(data):],
                [chunks[i] for i in data])
            for j, cid in enumerate(coding):
                assert np.array_equal(chunks[cid], want[j]), (
                    "lrc layer parity mismatch", cid)
    for _ in range(args.warmup):
        step()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    elapsed = time.perf_counter() - t0
    value = S * k * C * args.steps / GIB / elapsed
    print(json.dumps({
        "metric": "LRC layered encode GiB/s",
        "value": round(value, 2), "unit": "GiB/s", "n_gpus": 1,
        "steps": args.steps, "warmup": args.warmup,
        "ms_per_step": round(elapsed / args.steps * 1e3, 3),
        "higher_is_better": True, "scaling": "weak", "vs_baseline": None,
        "dtype": "u8", "data": "synthetic",
        "config": {"workload": ("LRC k=9 m=3 l=4 (nearest valid to "
                                "BASELINE configs[3]'s k=8 m=3 l=4, which "
                                "parse_kml rejects), 512 KiB chunks, "
                                "device-resident layered encode"),
                   "chunk_count": n, "stripes": S, "layers": len(layers),
                   "seed": hex(args.seed)},
    }))
    ctx.dbuf_free(d)
    ctx.close()


def bench_hostpath(args):
    """PCIe-inclusive plugin-path probe (single-stripe host-pointer calls,
    the drop-in path): reported separately from the device-resident metric
    per DESIGN.md §5 — never the headline value. --threads N measures N
    concurrent callers (the OSD's PG workers; ctypes releases the GIL, and
    the context round-robins its stream slots --streams wide)."""
    import ceph_amd
    k, m, C = args.k, args.m, args.chunk_bytes
    ctx = ceph_amd.EcContext(k, m, args.technique, device=0,
                             n_streams=args.streams)
    rng = np.random.default_rng(args.seed)
    data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
    ctx.encode_chunks(data)  # warm

    if args.threads > 1:
        from concurrent.futures import ThreadPoolExecutor
        datas = [[rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
                 for _ in range(args.threads)]

        outs = [None] * args.threads

        def worker(ti, stop):
            if outs[ti] is None:
                outs[ti] = ctx.encode_chunks(datas[ti])
            n = 0
            while time.perf_counter() < stop:
                ctx.encode_chunks(datas[ti], out=outs[ti])
                n += 1
            return n
        with ThreadPoolExecutor(args.threads) as ex:
            # warm every stream slot (pinned buffers allocate on first
            # use; rr round-robins calls over slots, so 2x threads calls
            # cover the pool) before the timed window opens
            list(ex.map(lambda ti: [ctx.encode_chunks(datas[ti])
                                    for _ in range(2)],
                        range(args.threads)))
            stop = time.perf_counter() + 8.0
            t0 = time.perf_counter()
            counts = list(ex.map(lambda ti: worker(ti, stop),
                                 range(args.threads)))
        dt = time.perf_counter() - t0
        print(json.dumps({
            "metric": "EC host-path (PCIe-inclusive) encode GiB/s",
            "value": round(sum(counts) * k * C / GIB / dt, 3),
            "unit": "GiB/s", "n_gpus": 1,
            "note": (f"{args.threads} concurrent host threads over "
                     f"{args.streams} stream slots; drop-in plugin path"),
            "config": {"k": k, "m": m, "chunk_bytes": C,
                       "threads": args.threads, "streams": args.streams,
                       "iters": sum(counts)},
        }))
        ctx.close()
        return
    par_out = ctx.encode_chunks(data)  # reusable parity buffers
    t0 = time.perf_counter()
    iters = 0
    while time.perf_counter() - t0 < 8.0:
        ctx.encode_chunks(data, out=par_out)
        iters += 1
    dt = time.perf_counter() - t0
    enc_gibs = iters * k * C / GIB / dt

    # decode leg: m erasures (the worst single-call repair), host pointers
    parity = ctx.encode_chunks(data)
    chunks = [d.copy() for d in data] + [p.copy() for p in parity]
    present = [i >= m for i in range(k + m)]  # first m data chunks erased
    ctx.decode_chunks(chunks, present)  # warm
    t0 = time.perf_counter()
    dits = 0
    while time.perf_counter() - t0 < 8.0:
        ctx.decode_chunks(chunks, present)
        dits += 1
    ddt = time.perf_counter() - t0
    for e in range(m):
        assert np.array_equal(chunks[e], data[e]), "hostpath decode mismatch"
    print(json.dumps({
        "metric": "EC host-path (PCIe-inclusive) encode GiB/s",
        "value": round(enc_gibs, 3), "unit": "GiB/s",
        "n_gpus": 1, "note": ("single-stripe ecx_encode_chunks_host incl. "
                              "H2D+D2H staging; drop-in plugin path, not "
                              "the device-resident metric"),
        "decode_gibs": round(dits * k * C / GIB / ddt, 3),
        "decode_erasures": m,
        "config": {"k": k, "m": m, "chunk_bytes": C, "iters": iters,
                   "decode_iters": dits},
    }))
    ctx.close()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--stripes", type=int, default=4096,
                    help="stripes per GPU (weak scaling)")
    ap.add_argument("--chunk-kib", type=int, default=1024)
    ap.add_argument("--k", type=int, default=8)
    ap.add_argument("--m", type=int, default=3)
    ap.add_argument("--technique", default="reed_sol_van")
    ap.add_argument("--packetsize", type=int, default=2048,
                    help="bitmatrix techniques only (jerasure packetsize)")
    ap.add_argument("--dry-run", action="store_true",
                    help="emit the per-rank run plan and exit (no GPU)")
    ap.add_argument("--erasures", type=int, default=3)
    ap.add_argument("--seed", type=lambda x: int(x, 0), default=0xEC)
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--no-selfcheck", action="store_true")
    ap.add_argument("--streams", type=int, default=2)
    ap.add_argument("--threads", type=int, default=1,
                    help="concurrent host-path callers (hostpath preset)")
    ap.add_argument("--config", choices=["rs83", "cauchy104", "mixed",
                                         "hostpath", "slices", "lrc"],
                    default="rs83",
                    help="BASELINE preset: rs83=configs[1] (default), "
                         "cauchy104=configs[2], mixed=configs[4] shape "
                         "sweep w/ latency histogram, hostpath=PCIe "
                         "plugin-path probe")
    args = ap.parse_args()
    args.chunk_bytes = args.chunk_kib * 1024
    if args.config == "cauchy104":
        args.k, args.m, args.technique, args.erasures = 10, 4, "cauchy", 4
        args.stripes = min(args.stripes, 3072)  # 14 chunks/stripe, ~42 GiB

    # distributed setup (torchrun provides RANK/WORLD_SIZE/LOCAL_RANK).
    # ORDER MATTERS: when torch is needed, its (bundled) HIP runtime must
    # initialise BEFORE ceph_amd loads /opt/rocm's — loading ours first
    # leaves torch.cuda seeing zero devices (two HIP runtimes in one
    # process). ceph_amd then resolves libamdhip64 by soname to the
    # already-loaded copy.
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    dist = None
    device = local_rank
    if args.dry_run:
        # SCALE-run readiness check (CPU-only, no torch/GPU): emit the
        # per-rank plan the 8-GPU driver launch would use so CI can
        # verify arg/env parsing and the rank->device map without
        # hardware.
        k, m, C, S = args.k, args.m, args.chunk_bytes, args.stripes
        print(json.dumps({
            "dry_run": True, "rank": rank, "world": world,
            "local_rank": local_rank, "device": local_rank,
            "backend": "nccl" if world > 1 else None,
            "config": args.config, "technique": args.technique,
            "k": k, "m": m, "chunk_bytes": C, "stripes_per_gpu": S,
            "buf_bytes": S * (k + m) * C, "steps": args.steps,
            "warmup": args.warmup, "scaling": "weak",
            "master_addr": os.environ.get("MASTER_ADDR"),
            "master_port": os.environ.get("MASTER_PORT")}))
        return
    if world > 1:
        import torch
        import torch.distributed as tdist
        # modulo so an N-rank run also works on fewer devices (single-box
        # smoke tests); identity on the 8-GPU node. ECX_BENCH_BACKEND=gloo
        # lets a multi-rank run share ONE GPU (NCCL refuses duplicate
        # devices) — used to validate the full distributed path end to
        # end on a 1-GPU box; the driver's 8-GPU run keeps nccl/RCCL.
        backend = os.environ.get("ECX_BENCH_BACKEND", "nccl")
        device = local_rank % max(1, torch.cuda.device_count())
        torch.cuda.set_device(device)
        tdist.init_process_group(backend)
        dist = tdist

    import ceph_amd

    if ceph_amd.device_count() < 1:
        print(json.dumps({"error": "no GPU visible; bench requires MI355X"}))
        sys.exit(1)
    device = device % ceph_amd.device_count()

    if args.config == "mixed":
        return bench_mixed(args, dist, device)
    if args.config == "hostpath":
        return bench_hostpath(args)
    if args.config == "slices":
        return bench_slices(args)
    if args.config == "lrc":
        return bench_lrc(args)

    k, m, C, S = args.k, args.m, args.chunk_bytes, args.stripes
    n = k + m
    buf_bytes = S * n * C
    seed = args.seed + rank

    ctx = ceph_amd.EcContext(k, m, args.technique, device=device,
                             n_streams=args.streams,
                             packetsize=args.packetsize)
    dptr = ctx.dbuf_alloc(buf_bytes)
    ctx.fill_random(dptr, buf_bytes, seed)
    ctx.sync()

    present_mask = (1 << n) - 1
    erased = sorted(np.random.default_rng(args.seed).choice(
        n, size=args.erasures, replace=False).tolist())
    for e in erased:
        present_mask &= ~(1 << e)

    def step(timed_accum=None, dec_accum=None):
        ctx.encode_batch(dptr, S, C)
        if timed_accum is not None:
            timed_accum.append(ctx.last_kernel_ms())
        ctx.decode_batch(dptr, S, C, present_mask)
        if dec_accum is not None:
            dec_accum.append(ctx.last_kernel_ms())
        ctx.sync()

    # parity self-check before measuring (rank 0)
    step()
    if rank == 0 and not args.no_selfcheck:
        parity_selfcheck(ctx, dptr, args, seed)

    for _ in range(args.warmup):
        step()
    if dist:
        dist.barrier()
        import torch
        torch.cuda.synchronize()

    enc_ms, dec_ms = [], []
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step(enc_ms, dec_ms)
    elapsed = time.perf_counter() - t0
    if dist:
        import torch
        torch.cuda.synchronize()
        dev = "cuda" if dist.get_backend() == "nccl" else "cpu"
        t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
        dist.barrier()

    if rank == 0:
        input_bytes_per_step = 2 * k * C * S  # encode + decode accounting
        value = world * input_bytes_per_step * args.steps / GIB / elapsed
        enc_kernel_ms = float(np.mean(enc_ms)) if enc_ms else None
        # encode kernel algorithmic traffic: read k*C*S + write m*C*S
        alg_bytes = (k + m) * C * S
        achieved = alg_bytes / (enc_kernel_ms * 1e-3) / 1e9 if enc_kernel_ms else None
        peak = 8000.0  # GB/s, MI355X HBM3E spec (MI355X_MICROARCH.md)
        line = {
            "metric": "EC encode+decode GiB/s",
            "value": round(value, 2),
            "unit": "GiB/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no absolute EC numbers
            "dtype": "u8",
            "data": "synthetic",
            "config": {
                "workload": (
                    f"{args.technique} k={k} m={m}, {C >> 10} KiB chunks, "
                    f"{S}-stripe batch encode+decode({len(erased)} erasures)"
                    ", device-resident "
                    + ("(BASELINE configs[2])" if args.config == "cauchy104"
                       else "(BASELINE configs[1]+[2])")),
                "k": k, "m": m, "chunk_bytes": C, "stripes_per_gpu": S,
                "technique": args.technique, "erasures": erased,
                "seed": hex(args.seed),
                "encode_gibs": round(world * k * C * S * args.steps / GIB /
                                     elapsed * 2, 2) if False else None,
            },
            "roofline": {
                "bound": "hbm",
                "achieved": round(achieved, 1) if achieved else None,
                "peak": peak,
                "unit": "GB/s",
                "frac": round(achieved / peak, 4) if achieved else None,
                "traffic": None,  # PMC traffic comes from rocprofv3 runs
                                  # committed under profiles/
                "kernel": "ec_gf_matmul_kernel (encode leg)",
                "kernel_ms": round(enc_kernel_ms, 4) if enc_kernel_ms else None,
                "alg_bytes_per_launch": alg_bytes,
                "decode_kernel_ms": (round(float(np.mean(dec_ms)), 4)
                                     if dec_ms else None),
                "decode_alg_bytes": (k + len(erased)) * C * S,
            },
            "cpu_baseline": (cpu_baseline(args)
                             if (world == 1 and not args.no_cpu_baseline
                                 and args.technique in
                                 ("reed_sol_van", "cauchy",
                                  "jerasure_reed_sol_van"))
                             else None),
        }
        del line["config"]["encode_gibs"]
        print(json.dumps(line))

    ctx.dbuf_free(dptr)
    ctx.close()
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
