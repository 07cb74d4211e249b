"""Multi-process CPU coverage of the N>1 bench path (gloo, world_size 2):
per-rank independent compaction jobs (oracle standing in for the GPU
engine on this GPU-less host), the RCCL-analog all-gather of emitted byte
counts, and the max-over-ranks elapsed reduction — the exact collective
shape bench.py uses with nccl on the 8x MI355X node."""
import os

import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import oracle
        from dbeel_amd.genruns import make_runs

        runs = make_runs(3, 400, 16, 64, overlap_frac=0.5,
                         tombstone_frac=0.1, seed=1000 + rank)
        data, index, n = oracle.compact(runs, keep_tombstones=False)

        counts = torch.tensor([len(data)], dtype=torch.int64)
        gathered = [torch.zeros_like(counts) for _ in range(world)]
        dist.barrier()
        dist.all_gather(gathered, counts)

        elapsed = torch.tensor([0.1 * (rank + 1)], dtype=torch.float64)
        dist.all_reduce(elapsed, op=dist.ReduceOp.MAX)

        results[rank] = {
            "counts": [int(g.item()) for g in gathered],
            "own": len(data),
            "max_elapsed": float(elapsed.item()),
            "n": n,
        }
    finally:
        dist.destroy_process_group()


def test_independent_jobs_allgather_two_ranks():
    world = 2
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29531
        procs = [
            ctx.Process(target=_worker, args=(r, world, port, results))
            for r in range(world)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(120)
            assert p.exitcode == 0
        res = dict(results)

    assert set(res) == {0, 1}
    # every rank sees every rank's emitted byte count, in rank order
    assert res[0]["counts"] == res[1]["counts"]
    assert res[0]["counts"][0] == res[0]["own"]
    assert res[0]["counts"][1] == res[1]["own"]
    # ranks used different seeds -> different jobs
    assert res[0]["own"] != res[1]["own"]
    # max-over-ranks elapsed
    assert res[0]["max_elapsed"] == pytest.approx(0.2)
    assert all(r["n"] > 0 for r in res.values())
