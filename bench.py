#!/usr/bin/env python3
"""Benchmark: dbeel SSTable compaction on MI355X.

Workload (BASELINE.json metric config — the 8-run merge the headline metric
is quoted on): config 3 = 8 runs x ~1 GiB, 32 B keys / 1 KiB values, 50% key
overlap, 5% tombstones, synthetic seeded runs (no network). A "step" = one
full compaction of the 8-run set with inputs already resident in HBM
(outputs land in HBM). The default line also carries `end_to_end`: the
PCIe-inclusive streamed-pinned-ingest rate measured beside the resident
metric (never as `value` — SURVEY.md §8d), plus `cpu_baseline` (the C
oracle restatement on the box's host cores). Other BASELINE shapes via
--workload cfg2/cfg4/cfg5.

Multi-GPU: the path shards as INDEPENDENT jobs — one
compaction job per GPU (weak scaling), with the only collective an RCCL
all-gather of per-rank emitted byte counts (8 x u64-scale) per step, per the
north star.

Usage:  python bench.py [--gpus N] [--steps K] [--warmup W] [--scale S]
For N>1 the driver launches via torch.distributed.run (one rank per GPU);
rank/device from RANK/LOCAL_RANK/WORLD_SIZE.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

HBM_PEAK_GBPS = 8000.0  # MI355X HBM3E spec peak, GB/s (MI355X_MICROARCH.md)


def _build_workload(scale: float, workload: str):
    from dbeel_amd.genruns import CONFIGS, make_runs

    if workload == "cfg5":
        from dbeel_amd.genruns import make_runs_varkey

        cfg = dict(CONFIGS["cfg5"])
        if scale != 1.0:
            cfg["entries_per_run"] = max(64, int(cfg["entries_per_run"] * scale))
        runs = make_runs_varkey(seed=0xDBEE1, **cfg)
        cfg["key_size"] = 128  # upper bound; keys are 8-128 B zipf
        return [runs], cfg
    if workload == "cfg4":
        cfg = dict(CONFIGS["cfg4_job"])
        if scale != 1.0:
            cfg["entries_per_run"] = max(64, int(cfg["entries_per_run"] * scale))
        jobs = [make_runs(seed=0xDBEE1 + 101 * j, **cfg) for j in range(8)]
        return jobs, cfg
    name = "cfg2" if workload == "cfg2" else "cfg3"
    cfg = dict(CONFIGS[name])
    if scale != 1.0:
        cfg["entries_per_run"] = max(64, int(cfg["entries_per_run"] * scale))
    runs = make_runs(seed=0xDBEE1, **cfg)
    return [runs], cfg


def make_workload(rank: int, scale: float, workload: str = "cfg3"):
    """cfg3: one 8-run x 1 GiB job per GPU (the metric config).
    cfg4: 8 independent 4-run x 256 MiB jobs per GPU (= BASELINE configs[3],
    64 jobs over 8 GPUs, one dbeel shard each). Returns (job_list, cfg).

    Every rank gets the SAME seeded synthetic content (weak scaling over
    independent jobs — identical per-rank replicas are statistically the
    same workload), so a box-local cache built by the first invocation
    serves all ranks of every later -N run in a SCALE sweep instead of
    re-generating GiBs per rank per run. Disable with
    DBEEL_BENCH_CACHE=off."""
    import numpy as np

    del rank  # content is rank-independent by design (see docstring)
    cache_dir = os.environ.get("DBEEL_BENCH_CACHE", "/tmp/dbeel_bench_cache")
    if cache_dir.lower() in ("off", "0", ""):
        return _build_workload(scale, workload)
    path = os.path.join(cache_dir, f"{workload}_{scale}.npz")
    _, cfg = (None, None)
    if os.path.exists(path):
        try:
            z = np.load(path, allow_pickle=True)
            counts = z["job_counts"]
            jobs = []
            k = 0
            for c in counts:
                jobs.append([(z[f"d{k + r}"], z[f"i{k + r}"])
                             for r in range(int(c))])
                k += int(c)
            cfg = json.loads(str(z["cfg"]))
            return jobs, cfg
        except Exception:
            pass  # corrupt/partial cache: rebuild
    jobs, cfg = _build_workload(scale, workload)
    try:
        import tempfile

        os.makedirs(cache_dir, exist_ok=True)
        arrs = {"job_counts": np.array([len(r) for r in jobs]),
                "cfg": np.array(json.dumps(cfg))}
        k = 0
        for runs in jobs:
            for d, i in runs:
                arrs[f"d{k}"] = d
                arrs[f"i{k}"] = i
                k += 1
        fd, tmp = tempfile.mkstemp(dir=cache_dir, suffix=".npz")
        os.close(fd)
        np.savez(tmp, **arrs)
        # np.savez appends .npz when missing; tmp already ends with it
        os.replace(tmp, path)
    except Exception:
        pass  # cache is best-effort
    return jobs, cfg


def aux_record_bytes(key_size: int) -> int:
    # aux tier (dbeel_gpu.hip AuxT): 16-B ts-less records for klen<=20,
    # 48-B with staged ts for klen<=32, 64-B with ts beyond
    return 16 if key_size <= 20 else (48 if key_size <= 32 else 64)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--scale", type=float, default=1.0,
                    help="scale entries_per_run (1.0 = full config 3)")
    ap.add_argument("--workload", choices=["cfg2", "cfg3", "cfg4", "cfg5"],
                    default="cfg3",
                    help="cfg2 = 4-run x 1M x 304 B (BASELINE configs[1]); "
                         "cfg3 = 8-run x 1 GiB metric config; cfg4 = 8 "
                         "independent 4-run x 256 MiB jobs per GPU; cfg5 = "
                         "16-run var-length msgpack keys + 4 KiB values")
    ap.add_argument("--skip-streamed", action="store_true",
                    help="skip the end-to-end streamed-ingest measurement")
    ap.add_argument("--keep-tombstones", action="store_true")
    ap.add_argument("--cpu-baseline-scale", type=float, default=1.0,
                    help="fraction of the workload timed on the host "
                         "(default: the whole workload, one single-threaded "
                         "oracle job per core)")
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    args = ap.parse_args()

    import torch

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    n_gpus = max(world, args.gpus)
    # one rank per GPU; tolerate rehearsal runs with more ranks than GPUs
    n_dev = torch.cuda.device_count() if torch.cuda.is_available() else 0
    device = local_rank % n_dev if n_dev else local_rank
    backend = os.environ.get("DBEEL_BENCH_BACKEND", "nccl")
    dist = None
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        dist.init_process_group(backend=backend)
        if n_dev:
            torch.cuda.set_device(device)

    import dbeel_amd

    keep = bool(args.keep_tombstones)
    job_runs, cfg = make_workload(rank, args.scale, args.workload)
    input_bytes = sum(
        d.nbytes + i.nbytes for runs in job_runs for d, i in runs
    )
    n_entries = sum(i.nbytes // 16 for runs in job_runs for d, i in runs)

    if len(job_runs) > 1:
        # independent jobs, one HIP stream each (one dbeel shard per job —
        # SURVEY.md §8e); ctypes releases the GIL during the blocking
        # engine calls. DBEEL_BENCH_BATCHED=1 runs them as ONE batched
        # launch set instead (dbeel_gpu_job_create_batched) — measured
        # equivalent on cfg4 (6.83 vs 6.69 ms: the work, not the launch
        # pattern, bounds this shape), kept for A/B.
        if os.environ.get("DBEEL_BENCH_BATCHED") == "1":
            jobs = [dbeel_amd.engine.BatchJob(job_runs, device=device)]
        else:
            jobs = [dbeel_amd.Job(runs, device=device) for runs in job_runs]
    else:
        jobs = [dbeel_amd.Job(job_runs[0], device=device)]

    pool = None
    if len(jobs) > 1:
        from concurrent.futures import ThreadPoolExecutor

        pool = ThreadPoolExecutor(max_workers=len(jobs))

    def run_all():
        if pool:
            results = list(pool.map(lambda j: j.run(keep), jobs))
        else:
            results = [jobs[0].run(keep)]
        tb = sum(r[0] for r in results)
        te = sum(r[1] for r in results)
        t_acc = None
        for _, _, t in results:
            if t_acc is None:
                t_acc = dict(t)
            else:
                for k in t_acc:
                    t_acc[k] += t[k]
        return tb, te, t_acc

    # Warmup (untimed)
    out_bytes = out_entries = 0
    for _ in range(args.warmup):
        out_bytes, out_entries, _ = run_all()

    def barrier_sync():
        if torch.cuda.is_available():
            torch.cuda.synchronize(device)
        if dist:
            dist.barrier()
        if torch.cuda.is_available():
            torch.cuda.synchronize(device)

    barrier_sync()
    t0 = time.perf_counter()
    tims = []
    for _ in range(args.steps):
        out_bytes, out_entries, t = run_all()
        tims.append(t)
        if dist:
            # the path's only collective: RCCL all-gather of emitted byte
            # counts over xGMI (north_star / SURVEY.md §5)
            dev = f"cuda:{device}" if backend == "nccl" else "cpu"
            counts = torch.tensor([out_bytes], dtype=torch.int64, device=dev)
            gathered = [torch.zeros_like(counts) for _ in range(world)]
            dist.all_gather(gathered, counts)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if dist:
        dev = f"cuda:{device}" if backend == "nccl" else "cpu"
        e = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())

    ms_per_step = elapsed * 1000.0 / args.steps
    total_input = input_bytes * n_gpus
    value = (total_input / 1e6) / (elapsed / args.steps)  # MB/s, whole job

    avg = {k: sum(t[k] for t in tims) / len(tims) for k in tims[0]}
    # dominant kernel by average time
    per_kernel = {
        "prep": avg["prep_ms"],
        "rank": avg["rank_ms"],
        "scan": avg["scan_ms"],
        "emit": avg["emit_ms"],
        "copy": avg["copy_ms"],
    }
    dom = max(per_kernel, key=per_kernel.get)
    out_index_bytes = out_entries * 16
    k = cfg["n_runs"]
    aux = aux_record_bytes(cfg["key_size"])
    algo = {
        # per-kernel algorithmic bytes (r02 pipeline)
        # prep: index rec + key bytes read; pfx (8) + aux written
        "prep": n_entries * (16 + cfg["key_size"] + 8 + aux),
        # corank: one pfx pass per pair side + cr written; rankreduce:
        # index + pfx + aux + cr read, 16-B rrec written
        "rank": n_entries * (2 * (k - 1) * 8 // max(k, 1) + 4 * (k - 1)
                             + 16 + 8 + aux + 4 * (k - 1) + 16),
        # two survivor scans: 16-B rrec read per pass, offsets written
        "scan": (2 * 16 + 8 + 4) * n_entries,
        # emit: rrec + offsets read, index rec + src_map written
        "emit": 28 * n_entries + 24 * out_entries,
        "copy": 2 * out_bytes + 24 * out_entries,
    }
    dom_ms = per_kernel[dom]
    achieved = (algo[dom] / 1e9) / (dom_ms / 1e3) if dom_ms > 0 else 0.0
    # PMC-measured per-launch HBM traffic (rocprofv3 --pmc FETCH_SIZE /
    # WRITE_SIZE in separate passes, gfx950 2x fetch correction calibrated
    # against k_copy's known byte counts — see profiles/r01_traffic_cfg3.json)
    traffic = None
    pipe_traffic = None
    import glob as _glob

    tcands = sorted(_glob.glob(os.path.join(
        os.path.dirname(os.path.abspath(__file__)),
        "profiles", "r*_traffic_cfg3.json")))
    tpath = tcands[-1] if tcands else None
    if args.scale == 1.0 and args.workload == "cfg3" and tpath:
        tj = json.load(open(tpath))
        kmap = tj.get("kernels", {})
        traffic = kmap.get(f"k_{dom}", {}).get("traffic_bytes")
        pipe_traffic = sum(v["traffic_bytes"] for v in kmap.values())
    roofline = {
        "bound": "hbm",
        "kernel": f"k_{dom}",
        "achieved": round(achieved, 1),
        "peak": HBM_PEAK_GBPS,
        "unit": "GB/s",
        "frac": round(achieved / HBM_PEAK_GBPS, 4),
        "traffic": traffic,
    }
    # whole-pipeline roofline (SURVEY.md §8d algorithmic B over the WALL
    # step time — per-job HIP-event spans double-count time-slicing when
    # multiple jobs share the chip, so ms_per_step is the honest basis)
    B = input_bytes + out_bytes + out_index_bytes
    pipe_gbps = (B / 1e9) / (ms_per_step / 1e3)
    roofline_pipeline = {
        "bound": "hbm",
        "achieved": round(pipe_gbps, 1),
        "peak": HBM_PEAK_GBPS,
        "unit": "GB/s",
        "frac": round(pipe_gbps / HBM_PEAK_GBPS, 4),
        "traffic": pipe_traffic,
    }

    # End-to-end streamed ingest (north_star: pinned host DRAM ->
    # hipMemcpyAsync chunks overlapped with the prepare kernel). Measured
    # beside the resident metric ("end-to-end MB/s incl.
    # PCIe streaming is also reported"); never `value`.
    end_to_end = None
    if (rank == 0 and n_gpus == 1 and not args.skip_streamed
            and len(jobs) == 1):
        from dbeel_amd.engine import pin_host, unpin_host

        bufs = [a for d, i in job_runs[0] for a in (d, i)]
        try:
            for a in bufs:
                pin_host(a)
            pinned = True
        except Exception:
            pinned = False  # fall back to pageable streaming
        try:
            jobs[0].ingest(job_runs[0])  # warmup
            jobs[0].run(keep)
            e2e_steps = 3
            ist_acc = None
            t0 = time.perf_counter()
            for _ in range(e2e_steps):
                ist = jobs[0].ingest(job_runs[0])
                jobs[0].run(keep)
                ist_acc = ist if ist_acc is None else {
                    k: ist_acc[k] + ist[k] for k in ist
                }
            e2e_s = time.perf_counter() - t0
            end_to_end = {
                "MBps": round((input_bytes / 1e6) / (e2e_s / e2e_steps), 1),
                "ingest_ms": round(ist_acc["ingest_ms"] / e2e_steps, 3),
                "h2d_copy_ms": round(ist_acc["copy_ms"] / e2e_steps, 3),
                "prep_hidden_in_ingest": True,
                "chunks": int(ist_acc["chunks"] // e2e_steps),
                "pinned": pinned,
                "steps": e2e_steps,
            }
        finally:
            if pinned:
                for a in bufs:
                    try:
                        unpin_host(a)
                    except Exception:
                        pass

    cpu_baseline = None
    if rank == 0 and n_gpus == 1 and not args.skip_cpu_baseline:
        import oracle

        bs = args.cpu_baseline_scale * args.scale
        cjobs, _ = make_workload(0, bs, args.workload)
        cbytes = sum(d.nbytes + i.nbytes for runs in cjobs for d, i in runs)
        # dbeel runs one single-threaded compaction per shard core
        # (SURVEY.md §8d): 1 core for a single job, one core per job for
        # the multi-job config
        cores = min(len(cjobs), os.cpu_count() or 1)
        c0 = time.perf_counter()
        if cores > 1:
            from concurrent.futures import ThreadPoolExecutor

            with ThreadPoolExecutor(max_workers=cores) as tp:
                list(tp.map(
                    lambda r: oracle.compact(r, keep_tombstones=keep), cjobs
                ))
        else:
            for cruns in cjobs:
                oracle.compact(cruns, keep_tombstones=keep)
        c1 = time.perf_counter()
        cpu_baseline = {
            "value": round((cbytes / 1e6) / (c1 - c0), 1),
            "unit": "MB/s",
            "cores": cores,
            "kind": "port",
            "sample": (
                f"{args.workload} shape at {bs:.2f} scale "
                f"({cbytes / 1e6:.0f} MB input, {c1 - c0:.1f}s on "
                f"{cores} core(s), one single-threaded job per core; "
                "C oracle restatement — Rust/glommio unbuildable here, "
                "BASELINE.md)"
            ),
        }

    if rank == 0:
        line = {
            "metric": "compaction_MBps_input",
            "value": round(value, 1),
            "unit": "MB/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # no published compaction number
            "dtype": "u8",
            "data": "synthetic",
            "config": {
                "workload": {"cfg2": "cfg2_4run_x_1M_x_304B",
                             "cfg3": "cfg3_8run_x_1GiB",
                             "cfg4": "cfg4_8jobs_4run_x_256MiB_per_gpu",
                             "cfg5": "cfg5_16run_varkey_4KiB"}[args.workload]
                + (f"_scale{args.scale}" if args.scale != 1.0 else ""),
                "n_runs": cfg["n_runs"],
                "entries_per_run": cfg["entries_per_run"],
                "key_size": cfg["key_size"],
                "value_size": cfg["value_size"],
                "overlap": cfg.get("overlap_frac", 0),
                "tombstones": cfg.get("tombstone_frac", 0),
                "keep_tombstones": keep,
                "parallelism": f"independent_jobs_x{n_gpus}",
            },
            "kernel_ms": {k: round(v, 3) for k, v in avg.items()},
            "out_bytes": out_bytes,
            "out_entries": out_entries,
            "roofline": roofline,
            "roofline_pipeline": roofline_pipeline,
            "end_to_end": end_to_end,
            "cpu_baseline": cpu_baseline,
        }
        print(json.dumps(line))

    for j in jobs:
        j.close()
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
