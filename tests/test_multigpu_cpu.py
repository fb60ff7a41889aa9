"""Multi-process coverage of the distributed path's protocol on CPU (gloo,
world_size=2): rows shard by hash(key) across ranks (the reference's
RepartitionExec::Hash analog), the shared watermark is an
all-reduce MAX (the Arc<Mutex> watermark of streaming_window.rs:210), and the
union of the ranks' emissions must equal the unsharded run bit-exactly —
exactly the invariant bench.py's N-GPU mode relies on. The per-rank operator
here is the CPU oracle (stand-in for the GPU op, which needs hardware); the
GPU end of the same protocol is exercised by
tests/test_gpu_parity.py::test_watermark_injection and bench.py --gpus N."""
import os

import numpy as np
import pytest
import torch.multiprocessing as mp

from oracle import pyoracle


def _rank_main(rank, world, port, q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import torch

    o = pyoracle.Oracle(1000, 0)
    results = []
    wm_trace = []
    for step in range(4):
        ts, kid, val = pyoracle.gen(7, 1_000_000, step * 50_000, 50_000, 64, 40)
        mine = (kid % world) == rank  # hash(key) -> rank shard
        o.push(ts[mine], kid[mine], val[mine])
        wm = torch.tensor([o.watermark], dtype=torch.int64)
        dist.all_reduce(wm, op=dist.ReduceOp.MAX)
        # the shared watermark (dz_window_op_advance_watermark analog): ranks
        # see identical time distributions, so the all-reduced MAX must equal
        # each rank's local watermark — asserted below
        wm_trace.append((int(wm.item()), o.watermark))
        out = o.fetch()
        results.append(out)
    o.finish()
    results.append(o.fetch())
    o.close()
    dist.destroy_process_group()
    flat = {}
    for f in ("key", "count", "min", "max", "avg", "window_start"):
        flat[f] = np.concatenate([r[f] for r in results])
    q.put((rank, wm_trace, {f: a for f, a in flat.items()}))


@pytest.mark.timeout(300)
def test_key_sharded_world2_union_equals_unsharded():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29511
    procs = [ctx.Process(target=_rank_main, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    outs = {}
    for _ in range(2):
        rank, wm_trace, flat = q.get(timeout=240)
        outs[rank] = (wm_trace, flat)
    for p in procs:
        p.join(60)
        assert p.exitcode == 0

    # watermark agreement: the all-reduced MAX equals each rank's local
    # watermark at every step (identical time distribution per shard)
    for rank in (0, 1):
        for wm_global, wm_local in outs[rank][0]:
            assert wm_global == wm_local

    # union of shards == unsharded oracle, group for group
    o = pyoracle.Oracle(1000, 0)
    for step in range(4):
        ts, kid, val = pyoracle.gen(7, 1_000_000, step * 50_000, 50_000, 64, 40)
        o.push(ts, kid, val)
    o.finish()
    ref = o.fetch()
    o.close()

    def group_map(flat):
        return {(int(flat["key"][i]), int(flat["window_start"][i])):
                (int(flat["count"][i]), float(flat["min"][i]),
                 float(flat["max"][i]), float(flat["avg"][i]))
                for i in range(len(flat["key"]))}

    merged = group_map(outs[0][1])
    m1 = group_map(outs[1][1])
    assert not (set(merged) & set(m1)), "shards must own disjoint groups"
    merged.update(m1)
    refmap = group_map(ref)
    assert merged == refmap, "sharded union must be bit-exact vs unsharded"
