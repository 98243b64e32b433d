"""Multi-process distributed tests on CPU (gloo, world_size=2): dictionary
delta sync between shards — the same code path the 8-GPU RCCL bench uses."""
import multiprocessing as mp
import os

import pytest


def _worker(rank: int, world: int, port: int, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
        import torch.distributed as dist
        dist.init_process_group("gloo", rank=rank, world_size=world)

        from deepflow_amd.gen import SpanGenConfig
        from deepflow_amd.gen.spans import gen_span_payload
        from deepflow_amd.ingest import L7IngestPipeline
        from deepflow_amd.parallel.dict_sync import DictSync
        from deepflow_amd.store.kg import KnowledgeGraphTable, default_platform

        # each rank ingests a different stream slice (different services)
        cfg = SpanGenConfig(n=120, seed=500 + rank, tag_cardinality=30,
                            n_services=6, n_ips=32, n_attrs=2)
        kg = KnowledgeGraphTable(capacity_pow2=1 << 10, device="cpu")
        kg.update(default_platform(cfg))
        pipe = L7IngestPipeline(device="cpu", segment_rows=1 << 9, kg=kg,
                               dict_capacity=1 << 12,
                               time_base_s=cfg.base_time_ns // 10**9)
        sync = DictSync(pipe.dict, device="cpu")
        pipe.ingest_frame_payload(gen_span_payload(cfg))
        merged = sync.sync_step()
        # second step: no new entries -> cheap path
        pipe.ingest_frame_payload(gen_span_payload(cfg))
        merged2 = sync.sync_step()

        other = 1 - rank
        # remote hydration: the other rank saw different tag values (its
        # seed differs); we must have its entries
        remote_entries = len(sync.remote[other])
        # verify a remote id hydrates
        any_ok = False
        for (dom, ident), s in list(sync.remote[other].items())[:5]:
            got = sync.hydrate_remote(other, dom, ident)
            any_ok = any_ok or (got == s)
        dist.barrier()
        dist.destroy_process_group()
        q.put((rank, merged, merged2, remote_entries, any_ok))
    except Exception as e:  # pragma: no cover
        import traceback
        q.put((rank, "ERR", traceback.format_exc(), None, None))


@pytest.mark.timeout(120)
def test_dict_sync_two_ranks():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29511
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=110) for _ in range(2)]
    for p in procs:
        p.join(timeout=30)
    for r in results:
        assert r[1] != "ERR", r[2]
    for rank, merged, merged2, remote_entries, any_ok in results:
        assert merged > 0, "first sync should merge remote entries"
        assert merged2 == 0, "steady state should exchange nothing"
        assert remote_entries > 0
        assert any_ok
