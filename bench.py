#!/usr/bin/env python3
"""bench.py — measures BASELINE.json's metric: rows/sec through
window()+filter() on the synthetic sensor stream.

Workload (config.workload): BASELINE cfg2 AS WRITTEN — the largest
single-GPU config the metric is quoted on: 10k utf8 "sensor_{k}" keys
(per-row device interning inside the timed region — the GroupValues::intern
cost the reference pays), 1s tumbling window, count/min/max/avg(reading)
+ filter(max > 113), f64 readings, synthetic sensor stream (seeded; spec in
DESIGN.md §Generator). A step = one push of --rows-per-step rows through the
operator (device-resident inputs) including triggered window emission; the
default run covers 768M rows (≥ cfg2's 100M; longer runs amortize warmup so
the steady state dominates). --key-kind dense measures the pre-densified
int-key variant (round-1's headline).

Contract: `python bench.py --gpus N --steps K --warmup W`. For N>1 the driver
launches one rank per GPU via torch.distributed.run (RCCL). N-GPU mode
measures WEAK scaling over N independent key partitions: each rank generates
its own rank-distinct key shard (rank-local dense ids over a rank-seeded
stream), the in-engine analog of the rows a RepartitionExec::Hash(group_by)
exchange would deliver to that rank — the hash exchange itself
is not in the timed path (each GPU owns its keys for the whole run). The only
collective is the shared-watermark all-reduce (MAX), exchanged before each
push. scaling=weak: per-rank rows fixed as N grows.

Rank 0 prints ONE JSON line. cpu_baseline: the CPU oracle (kind "port") timed
on this box's host cores on a bounded sample of the same workload.
"""
import argparse
import ctypes
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

HBM_PEAK = 8.0e12  # B/s, MI355X spec peak (MI355X_MICROARCH.md §Chip-level)
ALG_BYTES_PER_ROW = 20.0  # key id 4B + reading 8B + ts 8B


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=96)
    p.add_argument("--warmup", type=int, default=6)
    p.add_argument("--rows-per-step", type=int, default=8_000_000)
    p.add_argument("--keys", type=int, default=10_000)
    p.add_argument("--rows-per-ms", type=int, default=1000)
    p.add_argument("--window-ms", type=int, default=1000)
    p.add_argument("--slide-ms", type=int, default=0)
    p.add_argument("--seed", type=int, default=42)
    p.add_argument("--key-kind", choices=["dense", "utf8"], default="utf8",
                   help="utf8 (default) = raw 'sensor_{k}' string keys pushed "
                        "to the op and interned ON DEVICE inside the timed "
                        "region — BASELINE cfg2 as written; dense = "
                        "pre-densified int ids (the round-1 variant, kept "
                        "for comparison)")
    p.add_argument("--no-filter", action="store_true")
    p.add_argument("--cfg5", action="store_true",
                   help="BASELINE cfg5 pipeline: inner join on trip_id "
                        "(1M-trip build side) feeding the 1s tumbling window "
                        "group-by driver; value = probe rows/s through "
                        "join()+window(). Per-GPU shard; rows all match "
                        "(build side resident before the timed region).")
    p.add_argument("--trips", type=int, default=1_000_000)
    p.add_argument("--ingest", action="store_true",
                   help="on-wire pipeline: pre-generated newline-delimited "
                        "JSON bytes (device-resident) -> device JSON decode "
                        "-> device utf8 intern -> window()+filter(); value = "
                        "decoded rows/s (the f4 ingest row; config reports "
                        "bytes/row so bytes/s = value * bytes_per_row)")
    p.add_argument("--dist-backend", default=None,
                   help="torch.distributed backend override (default: nccl on GPU)")
    p.add_argument("--staged-push", action="store_true",
                   help="use the staging-copy push instead of the zero-copy "
                        "borrowed push (also the fallback for A/B runs "
                        "against engine builds without the borrowed entry)")
    p.add_argument("--skip-cpu-baseline", action="store_true")
    p.add_argument("--debug-steps", action="store_true",
                   help="print per-step wall times (push / poll split) to stderr")
    p.add_argument("--cpu-sample-rows", type=int, default=8_000_000)
    return p.parse_args()


def cpu_baseline(args):
    """Time the CPU oracle (the reference-semantics restatement, kind 'port')
    on a bounded sample, 8-ways key-sharded across threads (per-group row
    order preserved => identical results; shard prep excluded from timing)."""
    import threading
    from oracle import pyoracle
    n = args.cpu_sample_rows
    ncores = min(8, os.cpu_count() or 1)
    ts, kid, val = pyoracle.gen(args.seed, 1_000_000, 0, n, args.keys,
                                args.rows_per_ms)
    shard_of = kid % ncores
    shards = []
    for t in range(ncores):
        m = shard_of == t
        shards.append((np.ascontiguousarray(ts[m]),
                       np.ascontiguousarray(kid[m]),
                       np.ascontiguousarray(val[m])))
    batch = 1_000_000
    ops = [pyoracle.Oracle(args.window_ms, args.slide_ms) for _ in range(ncores)]

    def run(t):
        sts, skid, sval = shards[t]
        for lo in range(0, len(sts), batch):
            ops[t].push(sts[lo:lo + batch], skid[lo:lo + batch], sval[lo:lo + batch])
        ops[t].finish()
        ops[t].fetch()

    threads = [threading.Thread(target=run, args=(t,)) for t in range(ncores)]
    t0 = time.perf_counter()
    for th in threads:
        th.start()
    for th in threads:
        th.join()
    dt = time.perf_counter() - t0
    for o in ops:
        o.close()
    return {
        "value": n / dt,
        "unit": "rows/s",
        "cores": ncores,
        "kind": "port",
        "sample": f"{n} rows of the same workload, key-sharded over "
                  f"{ncores} host threads, large batches (best-CPU-effort; "
                  "the reference's own 32-row-batch mode is slower — DESIGN.md)",
    }


def read_traffic():
    """Measured HBM bytes per dominant-kernel launch, if a rocprofv3 PMC run
    has been committed (profiles/hbm_traffic.json: {kernel: bytes_per_launch},
    collected per MI355X_MICROARCH.md §HBM: separate --pmc pass, FETCH_SIZE
    read-side x2 correction applied at collection time)."""
    path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                        "profiles", "hbm_traffic.json")
    try:
        with open(path) as f:
            return json.load(f)
    except Exception:
        return None


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if world > 1:
        import torch
        import torch.distributed as dist
        backend = args.dist_backend or ("nccl" if torch.cuda.is_available() else "gloo")
        dist.init_process_group(backend=backend)
        if torch.cuda.is_available() and backend == "nccl":
            torch.cuda.set_device(local_rank)
    else:
        dist = None

    import denormalized_amd as dz
    from denormalized_amd import _lib

    device = local_rank
    if world > 1:
        import torch
        device = local_rank % max(1, torch.cuda.device_count())
    K, W = args.steps, args.warmup
    B = args.rows_per_step
    total_rows = (K + W) * B

    # pre-generate the whole stream into HBM (inputs resident when the timed
    # region starts). Each rank owns a disjoint key shard (weak scaling):
    # same time distribution, rank-distinct keys (ids are rank-local).
    if args.ingest and args.cfg5:
        raise SystemExit("--ingest and --cfg5 are separate workloads")
    utf8 = args.key_kind == "utf8" and not args.cfg5 and not args.ingest
    gseed = args.seed + 1000003 * rank  # rank-distinct draws per shard
    d_ts = dz.DeviceArray(device, total_rows * 8)
    d_kid = None if (utf8 or args.cfg5) else dz.DeviceArray(device,
                                                            total_rows * 4)
    d_trips = join = None
    if args.cfg5:
        # probe stream: trip ids (int64) from the same seeded draw; build
        # side (trip -> driver) fully resident before the timed region
        d_trips = dz.DeviceArray(device, total_rows * 8)
    d_vals = dz.DeviceArray(device, total_rows * 8)
    dz.generate(device, gseed, 1_000_000, 0, total_rows,
                args.trips if args.cfg5 else args.keys,
                args.rows_per_ms, d_ts.ptr,
                d_trips.ptr if args.cfg5 else None,
                d_kid.ptr if d_kid else None, d_vals.ptr)
    if args.cfg5:
        rng = np.random.default_rng(gseed)
        trips_h = np.arange(args.trips, dtype=np.int64)
        drivers_h = rng.integers(0, args.keys, args.trips).astype(np.int64)
        d_btr = dz.DeviceArray(device, trips_h.nbytes)
        d_btr.from_host(trips_h)
        d_bdr = dz.DeviceArray(device, drivers_h.nbytes)
        d_bdr.from_host(drivers_h)
        join = dz.JoinOp(device=device, n_trips_hint=args.trips)
        join.push_build(args.trips, d_btr.ptr, d_bdr.ptr)
        d_btr.free()
        d_bdr.free()
    d_offs = d_data = None
    decoder = None
    json_base = []
    if args.ingest:
        # pre-generate the on-wire JSON stream into HBM, one slice per step
        d_lens = dz.DeviceArray(device, B * 4)
        d_joffs = dz.DeviceArray(device, (B + 1) * 8)
        total_bytes = 0
        step_lens = []
        for s_ in range(K + W):
            dz.generate_json(device, gseed, 1_000_000, s_ * B, B, args.keys,
                             args.rows_per_ms, d_lens=d_lens.ptr)
            lens = d_lens.to_host(np.int32, B)
            json_base.append(total_bytes)
            step_lens.append(lens)
            total_bytes += int(lens.sum())
        d_data = dz.DeviceArray(device, total_bytes)
        L = _lib.lib()
        for s_ in range(K + W):
            offs = np.zeros(B + 1, np.int64)
            np.cumsum(step_lens[s_], out=offs[1:])
            step_lens[s_] = int(offs[-1])  # slice byte length
            d_joffs.from_host(offs)
            dz.generate_json(device, gseed, 1_000_000, s_ * B, B, args.keys,
                             args.rows_per_ms, d_offsets=d_joffs.ptr,
                             d_data=ctypes.c_void_p(
                                 d_data.ptr.value + json_base[s_]))
        json_lens = step_lens
        d_lens.free()
        d_joffs.free()
        decoder = dz.JsonDecoder(device=device)
        json_bytes_total = total_bytes
    step_data_base = []
    if utf8:
        # utf8 key column, per-step Arrow slices: lens on device, host cumsum
        # into per-step int32 offsets, then the device fill pass. All
        # pre-generation — the timed region starts with everything resident.
        d_lens = dz.DeviceArray(device, B * 4)
        d_offs = dz.DeviceArray(device, (K + W) * (B + 1) * 4)
        host_offs = []
        total_bytes = 0
        for s in range(K + W):
            dz.generate_utf8(device, gseed, s * B, B, args.keys,
                             d_lens=d_lens.ptr)
            lens = d_lens.to_host(np.int32, B)
            o = np.zeros(B + 1, np.int32)
            np.cumsum(lens, out=o[1:])
            host_offs.append(o)
            step_data_base.append(total_bytes)
            total_bytes += int(o[-1])
        d_data = dz.DeviceArray(device, total_bytes)
        L = _lib.lib()
        for s in range(K + W):
            optr = ctypes.c_void_p(d_offs.ptr.value + s * (B + 1) * 4)
            L.dz_memcpy_h2d(optr, host_offs[s].ctypes.data_as(ctypes.c_void_p),
                            (B + 1) * 4)
            host_offs[s] = None  # cap host memory at one step's offsets
            dz.generate_utf8(device, gseed, s * B, B, args.keys,
                             d_offsets=optr,
                             d_key_data=ctypes.c_void_p(
                                 d_data.ptr.value + step_data_base[s]))
        d_lens.free()
        del host_offs
    dz.synchronize(device)

    op = dz.WindowOp(length_ms=args.window_ms, slide_ms=args.slide_ms,
                     aggs=[("count", 0), ("min", 0), ("max", 0), ("avg", 0)],
                     key_kind=(_lib.KEY_UTF8 if (utf8 or args.ingest)
                               else _lib.KEY_DENSE_INT64),
                     n_keys_hint=args.keys, device=device)
    if not args.no_filter:
        op.set_filter("max", ">", 113.0)

    def do_push(step):
        off = step * B
        if args.ingest:
            # on-wire bytes -> decode -> intern -> window, all on device
            decoder.decode(ctypes.c_void_p(d_data.ptr.value + json_base[step]),
                           json_lens[step])
            nr, pts, pko, pkd, pv = decoder.batch()
            op.push_device_utf8(nr, pts, pko, pkd, pv)
            return
        if args.cfg5:
            # join -> window, all on device: probe the build table, feed the
            # matched (ts, driver kid, value) columns to the window op
            # zero-copy (the join double-buffers its output, covering the
            # window's one-step borrowed lifetime)
            join.push_probe(B,
                            ctypes.c_void_p(d_ts.ptr.value + off * 8),
                            ctypes.c_void_p(d_trips.ptr.value + off * 8),
                            ctypes.c_void_p(d_vals.ptr.value + off * 8))
            nm, pts, pkid, pval = join.matches()
            if nm:
                op.push_device(nm, pts, pkid, pval, borrowed=True)
            return
        if utf8:
            op.push_device_utf8(
                B,
                ctypes.c_void_p(d_ts.ptr.value + off * 8),
                ctypes.c_void_p(d_offs.ptr.value + step * (B + 1) * 4),
                ctypes.c_void_p(d_data.ptr.value + step_data_base[step]),
                ctypes.c_void_p(d_vals.ptr.value + off * 8))
        else:
            op.push_device(B,
                           ctypes.c_void_p(d_ts.ptr.value + off * 8),
                           ctypes.c_void_p(d_kid.ptr.value + off * 4),
                           ctypes.c_void_p(d_vals.ptr.value + off * 8),
                           borrowed=not args.staged_push)

    def push_step(step):
        if dist is not None:
            # exchange the (one-batch-lagged) local watermarks BEFORE the
            # push: advance_watermark flushes the deferred previous batch at
            # exactly the point the push would have processed it anyway, so
            # the multi-rank step keeps the single-rank pipelining; the
            # one-batch lag is the reference's own async watermark
            # propagation semantics (late rows re-emit their frame)
            import torch
            wm = torch.tensor([op.watermark], dtype=torch.int64)
            if torch.cuda.is_available() and dist.get_backend() == "nccl":
                wm = wm.cuda()
            dist.all_reduce(wm, op=dist.ReduceOp.MAX)
            if int(wm.item()) != -(2**63):  # unset on the very first step
                op.advance_watermark(int(wm.item()))
        # borrowed (zero-copy) push: the pre-generated stream stays resident
        # and untouched for the whole run, exactly the lifetime the borrowed
        # contract asks for
        do_push(step)
        emitted = 0
        # non-blocking zero-copy poll: emission is pipelined on the op's
        # worker pool and overlaps the next step's kernels; consuming the
        # batch = reading the op-owned Arrow-style buffers (the C contract:
        # each batch is read before the next poll invalidates it), exactly
        # as the reference's downstream operator would.
        for b in op.poll_iter(copy=False):
            emitted += b["n_rows"]
        return emitted

    emitted = 0
    import gc
    gc.collect()
    gc.disable()  # a gen-0 pass every ~7 steps showed up as 1-2 ms stalls
    for s in range(W):
        emitted += push_step(s)
    dz.synchronize(device)
    if dist is not None:
        dist.barrier()

    t0 = time.perf_counter()
    if args.debug_steps:
        marks = []
        for s in range(W, W + K):
            a = time.perf_counter()
            do_push(s)
            m = time.perf_counter()
            for bt in op.poll_iter(copy=False):
                emitted += bt["n_rows"]
            marks.append((m - a, time.perf_counter() - m))
        tf = time.perf_counter()
    else:
        for s in range(W, W + K):
            emitted += push_step(s)
    op.finish()
    if args.debug_steps:
        print(f"finish {(time.perf_counter()-tf)*1e3:.3f} ms", file=sys.stderr)
        for i, (pu, po) in enumerate(marks):
            print(f"step {i} push {pu*1e3:7.3f} ms poll {po*1e3:7.3f} ms",
                  file=sys.stderr)
    op.drain()
    emitted += sum(b["n_rows"] for b in op.poll_iter(copy=False))
    dz.synchronize(device)
    if dist is not None:
        dist.barrier()
    t1 = time.perf_counter()
    gc.enable()

    elapsed = t1 - t0
    if dist is not None:
        import torch
        e = torch.tensor([elapsed], dtype=torch.float64)
        if torch.cuda.is_available() and dist.get_backend() == "nccl":
            e = e.cuda()
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())

    stats = op.kernel_stats()

    if rank == 0:
        rows_timed = K * B * world  # whole-job rows through the timed region
        value = rows_timed / elapsed
        # roofline for the dominant DEVICE kernel (host-phase timers excluded):
        # achieved = the kernel's PMC-calibrated HBM bytes per launch (from
        # the committed rocprofv3 --pmc run, profiles/hbm_traffic.json) over
        # its HIP-event average launch time. Falls back to the algorithmic
        # estimate only when no PMC calibration exists for the kernel.
        dev = {k: v for k, v in stats.items() if not k.startswith("h_")}
        dom = max(dev, key=lambda k: dev[k]["total_ms"])
        d = stats[dom]
        launches_timed = d["launches"]
        avg_s = (d["total_ms"] / 1000.0) / max(1, d["launches"])
        rows_per_launch = (K + W) * B / max(1, d["launches"])
        traffic = None
        tmap = read_traffic()
        per_k = tmap.get("per_kernel_bytes_per_launch", {}) if tmap else {}
        if tmap:
            traffic = per_k.get(dom)
        if traffic:
            achieved = traffic / avg_s
            achieved_basis = "pmc"
        else:
            achieved = ALG_BYTES_PER_ROW * rows_per_launch / avg_s
            achieved_basis = "algorithmic"
        # whole-path view: PMC bytes moved per step (sum over the pipeline's
        # kernels, launches from this run) and algorithmic bytes per step,
        # each over the measured wall step time — the gap between the two
        # fracs IS the pipeline's traffic amplification.
        step_s = elapsed / K
        pmc_bytes_per_step = None
        if per_k:
            tot = 0.0
            for kname, kstat in dev.items():
                bpl = per_k.get(kname)
                if bpl and kname != "gen":  # gen runs pre-timed-region
                    tot += bpl * kstat["launches"] / (K + W)
            pmc_bytes_per_step = tot if tot > 0 else None
        path = {
            "alg_Bps": ALG_BYTES_PER_ROW * B / step_s,
            "alg_frac": ALG_BYTES_PER_ROW * B / step_s / HBM_PEAK,
        }
        if pmc_bytes_per_step:
            path["pmc_Bps"] = pmc_bytes_per_step / step_s
            path["pmc_frac"] = pmc_bytes_per_step / step_s / HBM_PEAK
            path["amplification"] = pmc_bytes_per_step / (ALG_BYTES_PER_ROW * B)
        out = {
            "metric": ("rows/sec from on-wire JSON bytes through "
                       "decode()+window()+filter()" if args.ingest else
                       "probe rows/sec through join()+window() on synthetic "
                       "rideshare stream" if args.cfg5 else
                       "rows/sec through window()+filter() on synthetic "
                       "sensor stream"),
            "value": value,
            "unit": "rows/s",
            "n_gpus": world,
            "steps": K,
            "warmup": W,
            "ms_per_step": elapsed / K * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "f64",
            "data": "synthetic",
            "config": {
                "workload": (f"f4 ingest: on-wire JSON "
                             f"({json_bytes_total / (K + W) / B:.1f} B/row) "
                             "-> device decode -> device intern -> window, "
                             if args.ingest else
                             "cfg5 (per-GPU shard): join(trip_id, "
                             f"{args.trips} trips)->window group-by driver, "
                             if args.cfg5 else
                             (("cfg2-utf8 (as written, device-interned "
                               "string keys): " if utf8 else
                               "cfg2 (dense-int key variant): ")
                              if (args.keys == 10_000 and not args.slide_ms
                                  and args.window_ms == 1000) else ""))
                            + f"{K * B / 1e6:.0f}M rows/GPU, "
                            + (f"{args.keys} utf8 'sensor_{{k}}' keys "
                               "(per-row device intern in the timed region), "
                               if utf8 else f"{args.keys} dense-int keys, ")
                            + f"{args.window_ms}ms "
                            + (f"sliding/{args.slide_ms}ms hop" if args.slide_ms
                               else "tumbling")
                            + ", count/min/max/avg"
                            + ("" if args.no_filter else " + filter(max>113)"),
                "rows_per_step": B,
                "keys_per_gpu": args.keys,
                "rows_per_ms": args.rows_per_ms,
                "parallelism": f"key-sharded dp{world}",
                "emitted_rows_rank0": emitted,
            },
            "roofline": {
                "bound": "hbm",
                "kernel": dom,
                "achieved": achieved,
                "achieved_basis": achieved_basis,
                "peak": HBM_PEAK,
                "unit": "B/s",
                "frac": achieved / HBM_PEAK,
                "traffic": traffic,
                "launches": launches_timed,
                "path": path,
                "kernel_ms_total": {k: round(v["total_ms"], 3) for k, v in stats.items()},
            },
            "cpu_baseline": None,
        }
        if not args.skip_cpu_baseline and world == 1:
            out["cpu_baseline"] = cpu_baseline(args)
            if utf8:
                out["cpu_baseline"]["sample"] += (
                    "; NB the CPU leg aggregates pre-densified int keys — "
                    "per-row string interning is EXCLUDED from the CPU "
                    "baseline but INCLUDED in the GPU value (conservative "
                    "for the GPU/CPU ratio)")
        print(json.dumps(out))
    op.close()
    if join is not None:
        join.close()
    if decoder is not None:
        decoder.close()
    for a in (d_ts, d_kid, d_vals, d_offs, d_data, d_trips):
        if a is not None:
            a.free()
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
