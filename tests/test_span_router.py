"""Data-plane span routing: every record lands on the
shard owning its agent; nothing is lost or duplicated; routed shards
produce globally-correct queries. gloo world 2 and 8 (the code path the
8-GPU RCCL bench runs with all_to_all_single instead)."""
import multiprocessing as mp
import os

import numpy as np
import pytest


def _scan(payload: bytes):
    import ctypes as ct
    from deepflow_amd.ops import native
    arr = np.frombuffer(payload, dtype=np.uint8)
    max_n = max(len(payload) // 8, 16)
    offs = np.zeros(max_n, dtype=np.uint32)
    lens = np.zeros(max_n, dtype=np.uint32)
    n = int(native.cpu().df_scan_offsets(
        arr.ctypes.data_as(ct.c_void_p), len(payload),
        offs.ctypes.data_as(ct.c_void_p),
        lens.ctypes.data_as(ct.c_void_p), max_n))
    return arr, offs[:n].copy(), lens[:n].copy()


def test_shard_of_matches_vtap_hash():
    from deepflow_amd.gen import SpanGenConfig
    from deepflow_amd.gen.spans import gen_span_dict, gen_span_payload
    from deepflow_amd.parallel.span_router import shard_of
    from deepflow_amd.ops.ref import mix64
    cfg = SpanGenConfig(n=200, seed=7, tag_cardinality=20, n_agents=16,
                        n_ips=32, n_services=4, n_attrs=1)
    arr, offs, lens = _scan(gen_span_payload(cfg))
    shards = shard_of(arr, offs, lens, world=8)
    for i in range(cfg.n):
        vtap = gen_span_dict(cfg, i)["base"]["vtap_id"]
        want = mix64((1 << 32) | vtap) % 8
        assert int(shards[i]) == want, i
    # a given agent's records all land on one shard
    by_vtap = {}
    for i in range(cfg.n):
        vtap = gen_span_dict(cfg, i)["base"]["vtap_id"]
        by_vtap.setdefault(vtap, set()).add(int(shards[i]))
    assert all(len(s) == 1 for s in by_vtap.values())


def _worker(rank: int, world: int, port: int, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from deepflow_amd.gen import SpanGenConfig
        from deepflow_amd.gen.spans import gen_span_payload
        from deepflow_amd.ingest import L7IngestPipeline
        from deepflow_amd.parallel.span_router import SpanRouter, shard_of
        # DIFFERENT agents mixed into every rank's incoming stream
        cfg = SpanGenConfig(n=150, seed=900 + rank, tag_cardinality=25,
                            n_agents=32, n_ips=32, n_services=4, n_attrs=1)
        payload = gen_span_payload(cfg)
        arr, offs, lens = _scan(payload)
        router = SpanRouter(device="cpu")
        pay2, offs2, lens2 = router.route_cpu(arr, offs, lens)
        # every received record belongs to THIS shard
        got_shards = shard_of(pay2, offs2, lens2, world)
        owned = all(int(s) == rank for s in got_shards)
        pipe = L7IngestPipeline(device="cpu", segment_rows=1 << 10,
                                dict_capacity=1 << 12,
                                time_base_s=cfg.base_time_ns // 10**9)
        n_in = pipe.ingest(pay2, offs2, lens2) if len(offs2) else 0
        # global conservation: sum of ingested == world * n
        import torch
        t = torch.tensor([n_in], dtype=torch.int64)
        dist.all_reduce(t)
        total = int(t.item())
        # distributed query over the routed shards
        from deepflow_amd.query.engine import QueryEngine
        eng = QueryEngine(pipe, device="cpu")
        r = eng.query("SELECT Count(*) AS c FROM l7_flow_log")
        local_count = r["values"][0][0] if r["values"] else 0
        t2 = torch.tensor([local_count], dtype=torch.int64)
        dist.all_reduce(t2)
        dist.barrier()
        dist.destroy_process_group()
        q.put((rank, owned, n_in, total, int(t2.item())))
    except Exception:
        import traceback
        q.put((rank, "ERR", traceback.format_exc(), None, None))


@pytest.mark.timeout(180)
@pytest.mark.parametrize("world,port", [(2, 29611), (8, 29617)])
def test_route_exchange_gloo(world, port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=170) for _ in range(world)]
    for p in procs:
        p.join(timeout=30)
    for r in results:
        assert r[1] != "ERR", r[2]
    for rank, owned, n_in, total, qtotal in results:
        assert owned, f"rank {rank} received records it does not own"
        assert total == world * 150       # nothing lost or duplicated
        assert qtotal == world * 150      # and queryable after routing


@pytest.mark.gpu
@pytest.mark.parametrize("transport", ["torchdist", "rccl_direct"])
def test_route_gpu_world1(transport, monkeypatch):
    """GPU routing path on a single rank (RCCL pg, all_to_all_single with
    world=1): packed gather + exchange must reproduce the input records,
    and the routed batch must ingest identically to the unrouted one."""
    import torch
    import torch.distributed as dist
    assert torch.cuda.is_available()
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29741")
    if transport == "rccl_direct":
        monkeypatch.setenv("DF_RCCL_DIRECT", "1")
    else:
        monkeypatch.delenv("DF_RCCL_DIRECT", raising=False)
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        from deepflow_amd.gen import SpanGenConfig
        from deepflow_amd.gen.spans import gen_span_payload
        from deepflow_amd.ingest import L7IngestPipeline
        from deepflow_amd.parallel.span_router import SpanRouter
        cfg = SpanGenConfig(n=3000, seed=31, tag_cardinality=100,
                            n_agents=16, n_ips=64, n_services=8, n_attrs=2)
        arr, offs, lens = _scan(gen_span_payload(cfg))
        pay_t = torch.from_numpy(arr.copy()).cuda()
        offs_t = torch.from_numpy(offs.view(np.int32).copy()).cuda()
        lens_t = torch.from_numpy(lens.view(np.int32).copy()).cuda()
        router = SpanRouter(device="cuda")
        pay2, offs2, lens2 = router.route_gpu(pay_t, offs_t, lens_t,
                                              arr, offs, lens)
        assert int(lens2.sum()) == int(lens.sum())
        assert lens2.numel() == len(lens)
        # routed records ingest identically (same multiset of records)
        p1 = L7IngestPipeline(device="cuda", segment_rows=1 << 12,
                              dict_capacity=1 << 12,
                              time_base_s=cfg.base_time_ns // 10**9)
        p1.ingest_device(pay_t, offs_t, lens_t, arr)
        p2 = L7IngestPipeline(device="cuda", segment_rows=1 << 12,
                              dict_capacity=1 << 12,
                              time_base_s=cfg.base_time_ns // 10**9)
        p2.ingest_device(pay2, offs2, lens2, pay2)
        torch.cuda.synchronize()
        from deepflow_amd.query.engine import QueryEngine
        q = ("SELECT request_resource, Count(*) AS c FROM l7_flow_log "
             "GROUP BY request_resource ORDER BY c DESC, request_resource "
             "LIMIT 20")
        assert QueryEngine(p1, device="cuda").query(q) == \
            QueryEngine(p2, device="cuda").query(q)
    finally:
        from deepflow_amd.parallel import rccl as _rc
        if _rc._comm is not None:
            _rc._comm.close()
            _rc._comm = None
        dist.destroy_process_group()
