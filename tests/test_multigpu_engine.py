"""Multi-rank ENGINE coverage on one GPU (gloo, world_size=2): two processes
each own a dz_window_op on device 0 over a disjoint hash(key) shard of ONE
stream, exchange the shared watermark through an all-reduce MAX injected via
dz_window_op_advance_watermark before every push (bench.py's N-GPU step
shape), and the union of their emissions must equal the unsharded engine run
bit-exactly. This is the same protocol tests/test_multigpu_cpu.py covers with
oracle stand-ins — here the ENGINE's own multi-rank code path executes, so an
8-GPU scale run exercises nothing untested."""
import os

import numpy as np
import pytest
import torch.multiprocessing as mp

import __graft_entry__ as graft
from oracle import pyoracle

pytestmark = pytest.mark.gpu

STEPS = 4
ROWS = 60_000
KEYS = 96


def _rank_main(rank, world, port, q):
    import torch
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from denormalized_amd import WindowOp, _lib

    op = WindowOp(length_ms=1000, key_kind=_lib.KEY_INT64, n_keys_hint=KEYS,
                  device=0)
    results = []
    wm_trace = []
    for step in range(STEPS):
        # the one-batch-lagged watermark exchange BEFORE the push, exactly
        # like bench.py's push_step: advance_watermark flushes the deferred
        # batch at the point a push would have processed it anyway
        wm = torch.tensor([op.watermark], dtype=torch.int64)
        dist.all_reduce(wm, op=dist.ReduceOp.MAX)
        if int(wm.item()) != -(2 ** 63):
            op.advance_watermark(int(wm.item()))
        ts, kid, val = pyoracle.gen(11, 1_000_000, step * ROWS, ROWS, KEYS, 40)
        mine = (kid % world) == rank  # hash(key) -> rank shard
        op.push(ts[mine], kid[mine], val[mine])
        results += op.poll_all()
        wm_trace.append(op.watermark)
    op.finish()
    results += op.poll_all()
    op.close()
    dist.destroy_process_group()
    flat = {}
    for f in ("key", "count", "min", "max", "avg", "window_start"):
        flat[f] = np.concatenate([np.asarray(r[f]) for r in results]) \
            if results else np.zeros(0)
    q.put((rank, wm_trace, flat))


@pytest.mark.timeout(300)
def test_engine_key_sharded_world2_union_equals_unsharded():
    graft.build()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank_main, args=(r, 2, 29517, q))
             for r in range(2)]
    for p in procs:
        p.start()
    outs = {}
    for _ in range(2):
        rank, wm_trace, flat = q.get(timeout=240)
        outs[rank] = (wm_trace, flat)
    for p in procs:
        p.join(60)
        assert p.exitcode == 0

    # both ranks see identical time distributions => same final watermark
    assert outs[0][0][-1] == outs[1][0][-1]

    # unsharded ENGINE run over the full stream
    from denormalized_amd import WindowOp, _lib
    op = WindowOp(length_ms=1000, key_kind=_lib.KEY_INT64, n_keys_hint=KEYS,
                  device=0)
    for step in range(STEPS):
        ts, kid, val = pyoracle.gen(11, 1_000_000, step * ROWS, ROWS, KEYS, 40)
        op.push(ts, kid, val)
    op.finish()
    ref_rows = op.poll_all()
    op.close()

    def group_map(flat):
        return {(int(flat["key"][i]), int(flat["window_start"][i])):
                (int(flat["count"][i]), float(flat["min"][i]),
                 float(flat["max"][i]), float(flat["avg"][i]))
                for i in range(len(flat["key"]))}

    merged = group_map(outs[0][1])
    m1 = group_map(outs[1][1])
    assert merged and m1
    assert not (set(merged) & set(m1)), "shards must own disjoint groups"
    merged.update(m1)
    ref = {}
    for f in ("key", "count", "min", "max", "avg", "window_start"):
        ref[f] = np.concatenate([np.asarray(r[f]) for r in ref_rows])
    assert merged == group_map(ref), "sharded union must be bit-exact"
