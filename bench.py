#!/usr/bin/env python3
"""bench.py — KNN queries/sec on the BASELINE workload.

Workload (BASELINE.json configs[1]): 10M rows x 768-dim f32, brute-force
cosine K=10, single query at a time, per GPU. A "step" is ONE query over the
whole (sharded) corpus. Scaling is WEAK: each rank holds its own 10M-row
shard, so the corpus grows with N while per-GPU work per query is fixed
(BASELINE configs[4]-style sharding; rows_total = 10M * N).

N>1 protocol (one process per GPU, launched by torch.distributed.run):
every rank scans its shard (sdbv_knn_bruteforce), all ranks all-gather the
per-shard top-K (K*(f64,i64) per rank — latency-bound over xGMI via RCCL),
rank 0 merges with the reference tie-break (dist total_cmp asc, id asc).

Output: ONE JSON line from rank 0 (see the driver contract).
"""
import argparse
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

HBM_PEAK_GBS = 8000.0  # MI355X HBM3E spec peak (MI355X_MICROARCH.md)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=40)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--rows", type=int, default=10_000_000,
                    help="rows per GPU shard (weak scaling)")
    ap.add_argument("--dim", type=int, default=768)
    ap.add_argument("--k", type=int, default=10)
    ap.add_argument("--metric", default="cosine")
    ap.add_argument("--batch", type=int, default=0,
                    help=">0: run BASELINE configs[3] (batched MFMA path); a "
                         "step is one batch of this many queries")
    ap.add_argument("--hnsw", action="store_true",
                    help="run BASELINE configs[2]: HNSW index (M=16, ef=64) "
                         "K=10 single query; index built at bench start "
                         "(parallel host build, outside the timed region)")
    ap.add_argument("--ef", type=int, default=64)
    ap.add_argument("--hnsw-perhop", action="store_true",
                    help="HNSW mode: use the per-hop gather path instead of "
                         "the persistent kernel for single queries")
    ap.add_argument("--host-build", action="store_true",
                    help="HNSW mode: host chunked snapshot build instead of "
                         "the GPU-accelerated one")
    ap.add_argument("--chunk", type=int, default=8192,
                    help="HNSW snapshot build chunk size")
    ap.add_argument("--seed", type=lambda x: int(x, 0), default=0x5DB1)
    ap.add_argument("--cpu-sample-rows", type=int, default=10_000_000,
                    help="row bound for the cpu_baseline leg (defaults to "
                         "the full workload at the default size: no "
                         "extrapolation)")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    import torch

    if world == 1 and args.gpus > 1:
        # Not under torchrun: self-launch one rank per GPU, or fail loudly.
        # NEVER fall through to single-rank work reporting n_gpus > 1
        # (VERDICT r01 weak #3).
        ndev = torch.cuda.device_count() if torch.cuda.is_available() else 0
        if ndev < args.gpus:
            print(f"bench.py: --gpus {args.gpus} requested but only {ndev} "
                  "visible GPU(s) and WORLD_SIZE is unset; refusing to run "
                  "single-rank work as if it were multi-GPU", file=sys.stderr)
            sys.exit(2)
        import socket
        import subprocess
        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            port = s.getsockname()[1]
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={args.gpus}", "--master-addr", "127.0.0.1",
               f"--master-port={port}", os.path.abspath(__file__)]
        cmd += sys.argv[1:]
        sys.exit(subprocess.call(cmd))

    n_gpus = world
    dist = None
    if world > 1:
        import torch.distributed as tdist
        dist = tdist
        torch.cuda.set_device(local_rank)
        dist.init_process_group("nccl")

    import surrealdb_amd
    from surrealdb_amd.shard import merge_topk
    from surrealdb_amd.synth import gen_f32

    ctx = surrealdb_amd.Context(device=local_rank)

    # --- stage this rank's shard (one-time, outside the timed region) ---
    rows = args.rows
    row_offset = rank * rows
    t0 = time.perf_counter()
    hnsw_index = None
    if args.hnsw:
        # HNSW does not shard (sequential graph traversal):
        # replicas only — each rank builds/holds a replica of its shard.
        def mark(what):
            print(f"[bench hnsw] {what} at +{time.perf_counter()-t0:.0f}s",
                  file=sys.stderr, flush=True)
        from surrealdb_amd.synth import gen_f32 as _gen
        pts = _gen(args.seed, row_offset, rows, args.dim)
        mark("generated")
        hnsw_index = ctx.hnsw_create(args.dim, metric=args.metric, m=16,
                                     m0=32, efc=150, seed=args.seed)
        # chunked snapshot build (chunk/n <= 0.4% at bench scales — quality
        # contract in DESIGN §9); GPU-accelerated by default: per-chunk
        # efc-searches run as one persistent-kernel launch
        if args.host_build:
            hnsw_index.insert_batch_snapshot(pts, chunk=args.chunk,
                                             nthreads=os.cpu_count())
        else:
            hnsw_index.insert_batch_snapshot_gpu(pts, chunk=args.chunk,
                                                 nthreads=os.cpu_count())
        mark("built")
        del pts
        hnsw_index.finalize(1)
        mark("finalized")
    else:
        ctx.stage_synthetic(1, rows, args.dim, metric=args.metric,
                            seed=args.seed, row_offset=row_offset,
                            id_base=row_offset)
    stage_s = time.perf_counter() - t0

    nq = (args.steps + args.warmup) * max(args.batch, 1)
    queries = gen_f32(0xBEEF, 0, min(nq, 4096), args.dim)

    device = torch.device(f"cuda:{local_rank}")

    def one_batch(si):
        """One step of the batched path: args.batch concurrent queries."""
        b = args.batch
        q0 = (si * b) % max(len(queries) - b, 1)
        ids, dists = ctx.knn_batch(1, queries[q0:q0 + b], args.k)
        if world > 1:
            flat = torch.empty(b * args.k, 2, dtype=torch.float64,
                               device=device)
            flat[:, 0] = torch.from_numpy(dists.reshape(-1).copy()).to(device)
            flat[:, 1] = torch.from_numpy(
                ids.reshape(-1).view(np.float64).copy()).to(device)
            gathered = [torch.empty_like(flat) for _ in range(world)]
            dist.all_gather(gathered, flat)
            if rank == 0:
                g = [t.cpu().numpy().reshape(b, args.k, 2) for t in gathered]
                out = []
                for j in range(b):
                    out.append(merge_topk(
                        [x[j, :, 1].copy().view(np.uint64) for x in g],
                        [x[j, :, 0] for x in g], args.k))
                return out
            return None
        return ids, dists

    def one_hnsw_batch(si):
        """One step of the batched HNSW path: args.batch concurrent queries
        on the persistent in-kernel search (one workgroup per query)."""
        b = args.batch
        q0 = (si * b) % max(len(queries) - b, 1)
        ids, dists, ns = hnsw_index.knn_search_batch(queries[q0:q0 + b],
                                                     args.k, args.ef)
        if world > 1:
            # per-query merge across shard replicas (same protocol as the
            # brute-force batch path; HNSW shards queries over replicas of
            # row shards)
            flat = torch.empty(b * args.k, 2, dtype=torch.float64,
                               device=device)
            gids = ids.astype(np.uint64) + row_offset
            for j in range(b):
                gids[j, ns[j]:] = np.iinfo(np.uint64).max
                dists[j, ns[j]:] = np.inf
            flat[:, 0] = torch.from_numpy(dists.reshape(-1).copy()).to(device)
            flat[:, 1] = torch.from_numpy(
                gids.reshape(-1).view(np.float64).copy()).to(device)
            gathered = [torch.empty_like(flat) for _ in range(world)]
            dist.all_gather(gathered, flat)
            if rank == 0:
                g = [t.cpu().numpy().reshape(b, args.k, 2) for t in gathered]
                return [merge_topk(
                    [x[j, :, 1].copy().view(np.uint64) for x in g],
                    [x[j, :, 0] for x in g], args.k) for j in range(b)]
            return None
        return ids, dists

    def one_query(qi):
        if args.batch > 0:
            if hnsw_index is not None:
                return one_hnsw_batch(qi)
            return one_batch(qi)
        if hnsw_index is not None:
            q = queries[qi % len(queries)]
            # persistent kernel caps ef at 512 (LDS queue); larger ef takes
            # the per-hop path
            if args.hnsw_perhop or args.ef > 512:
                ids, dists = hnsw_index.knn_search(q, args.k, args.ef)
            else:
                # persistent kernel (whole best-first loop in ONE launch —
                # same exact results, validated in tests/test_gpu_hnsw.py)
                bids, bdists, bns = hnsw_index.knn_search_batch(
                    q.reshape(1, -1), args.k, args.ef)
                ids, dists = bids[0][:bns[0]], bdists[0][:bns[0]]
            ids = ids + row_offset  # shard-local ordinals -> global ids
        else:
            ids, dists = ctx.knn_bruteforce(1, queries[qi % len(queries)],
                                            args.k)
        if world > 1:
            pad = args.k - len(ids)
            if pad:
                ids = np.concatenate(
                    [ids, np.full(pad, np.iinfo(np.uint64).max, np.uint64)])
                dists = np.concatenate([dists, np.full(pad, np.inf)])
            local = torch.empty(args.k, 2, dtype=torch.float64, device=device)
            local[:, 0] = torch.from_numpy(dists.copy()).to(device)
            local[:, 1] = torch.from_numpy(
                ids.view(np.float64).copy()).to(device)
            gathered = [torch.empty_like(local) for _ in range(world)]
            dist.all_gather(gathered, local)
            if rank == 0:
                g = [t.cpu().numpy() for t in gathered]
                return merge_topk([x[:, 1].copy().view(np.uint64) for x in g],
                                  [x[:, 0] for x in g], args.k)
            return None
        return ids, dists

    # --- warmup ---
    for i in range(args.warmup):
        one_query(i)

    # hnsw-batch roofline calibration: the persistent kernel does not count
    # its gathered rows, but it visits exactly the per-hop path's set (same
    # best-first queue), so sample the per-hop gather count per query here,
    # outside the timed region.
    hnsw_gather_avg = None
    if args.hnsw and not args.hnsw_perhop:
        tot = 0
        nsample = min(32, len(queries))
        for j in range(nsample):
            hnsw_index.knn_search(queries[j], args.k, args.ef)
            tot += ctx.stats()["last_rows_scanned"]
        hnsw_gather_avg = tot / nsample

    # --- timed region: EXACTLY args.steps steps ---
    if dist:
        dist.barrier()
    torch.cuda.synchronize(device) if torch.cuda.is_available() else None
    step_times = []
    scan_ms_acc = 0.0
    rows_scanned_acc = 0
    t_begin = time.perf_counter()
    for i in range(args.steps):
        ts = time.perf_counter()
        one_query(args.warmup + i)
        step_times.append(time.perf_counter() - ts)
        st = ctx.stats()
        scan_ms_acc += st["last_scan_kernel_ms"]
        rows_scanned_acc += st["last_rows_scanned"]
    torch.cuda.synchronize(device) if torch.cuda.is_available() else None
    if dist:
        dist.barrier()
    t_total = time.perf_counter() - t_begin

    # max over ranks
    if dist:
        tt = torch.tensor([t_total], dtype=torch.float64, device=device)
        dist.all_reduce(tt, op=dist.ReduceOp.MAX)
        t_total = float(tt.item())

    if rank != 0:
        return

    queries_per_step = max(args.batch, 1)
    qps = args.steps * queries_per_step / t_total
    p50_ms = float(np.percentile(np.array(step_times) * 1e3, 50))
    p95_ms = float(np.percentile(np.array(step_times) * 1e3, 95))
    # early echo so a timeout during the (post-region) baseline leg still
    # leaves the core measurement on record
    print(f"[bench] value={qps:.3f} q/s p50={p50_ms:.3f}ms "
          f"p95={p95_ms:.3f}ms", file=sys.stderr, flush=True)

    scan_ms_avg = scan_ms_acc / args.steps
    traffic = os.environ.get("SDBV_TRAFFIC_BYTES_PER_LAUNCH")
    if args.batch > 0 and not args.hnsw:
        # dominant kernel = the f32 MFMA GEMM; stats.last_scan_kernel_ms is
        # the summed sgemm time of one step's chunks
        alg_flop = 2.0 * args.batch * rows * args.dim
        achieved_tf = alg_flop / (scan_ms_avg * 1e-3) / 1e12
        F32_MFMA_PEAK_TF = 157.3  # gfx950 f32-input MFMA dense peak
        roofline = {
            "bound": "mfma",
            "achieved": round(achieved_tf, 1),
            "peak": F32_MFMA_PEAK_TF,
            "unit": "TFLOP/s",
            "frac": round(achieved_tf / F32_MFMA_PEAK_TF, 4),
            "traffic": float(traffic) if traffic else None,
        }
    elif args.hnsw:
        # HNSW is latency/gather-bound: achieved = gathered rows
        # x row bytes over the GPU search sections. Per-hop mode counts its
        # gathers live; the persistent kernel visits the identical set, so
        # its count comes from the pre-timed-region per-hop calibration.
        if hnsw_gather_avg is not None:
            alg_bytes = hnsw_gather_avg * max(args.batch, 1) * args.dim * 4
        else:
            alg_bytes = (rows_scanned_acc / args.steps) * args.dim * 4
        achieved_gbs = alg_bytes / (scan_ms_avg * 1e-3) / 1e9
        roofline = {
            "bound": "hbm",
            "achieved": round(achieved_gbs, 1),
            "peak": HBM_PEAK_GBS,
            "unit": "GB/s",
            "frac": round(achieved_gbs / HBM_PEAK_GBS, 4),
            "traffic": float(traffic) if traffic else None,
        }
    else:
        # dominant kernel = the distance scan; algorithmic bytes per launch =
        # rows * d * 4 B (corpus read)
        alg_bytes = rows * args.dim * 4
        achieved_gbs = alg_bytes / (scan_ms_avg * 1e-3) / 1e9
        roofline = {
            "bound": "hbm",
            "achieved": round(achieved_gbs, 1),
            "peak": HBM_PEAK_GBS,
            "unit": "GB/s",
            "frac": round(achieved_gbs / HBM_PEAK_GBS, 4),
            "traffic": float(traffic) if traffic else None,
        }

    # --- cpu_baseline: the oracle (kind "port") on host cores, rank0/N=1 ---
    cpu_baseline = None
    if args.hnsw and world == 1 and not args.no_cpu_baseline:
        # oracle HNSW search over the EXACT graph the GPU searched: the
        # product graph exports layer by layer into orc_hnsw_import (the
        # oracle stays the checker/baseline — it never feeds the product
        # path). Search results are bit-identical by construction
        # (tests/test_hnsw_product.py::test_oracle_import_searches_...).
        import oracle
        layers = [hnsw_index.layer_csr(l)
                  for l in range(hnsw_index.num_layers())]
        og = oracle.Hnsw.import_graph(
            args.dim, args.metric, 16, 32, 150, hnsw_index.vecs_view(),
            hnsw_index.enter_point(), layers)
        del layers
        if args.batch > 0:
            # throughput config: CPU gets all cores, like the GPU gets all
            # its workgroups
            from concurrent.futures import ThreadPoolExecutor
            cores = os.cpu_count()
            nq_base = min(32 * cores, 4096)
            qlist = [queries[i % len(queries)] for i in range(nq_base)]
            with ThreadPoolExecutor(cores) as ex:  # ctypes drops the GIL
                list(ex.map(lambda q: og.search(q, args.k, args.ef),
                            qlist[:cores]))  # warmup
                tcs = time.perf_counter()
                list(ex.map(lambda q: og.search(q, args.k, args.ef), qlist))
                t_cpu = time.perf_counter() - tcs
            sample = (f"{nq_base} queries (ef={args.ef}) on the exported "
                      f"product graph ({rows} rows), oracle orc_hnsw_search "
                      f"across {cores} threads")
        else:
            # latency config (the BASELINE wording: SINGLE query): the CPU
            # leg must be sequential single queries like the GPU side —
            # threads here would compare CPU throughput against GPU latency
            cores = 1
            nq_base = 256
            for i in range(16):
                og.search(queries[i % len(queries)], args.k, args.ef)
            tcs = time.perf_counter()
            for i in range(nq_base):
                og.search(queries[i % len(queries)], args.k, args.ef)
            t_cpu = time.perf_counter() - tcs
            sample = (f"{nq_base} sequential queries (ef={args.ef}) on the "
                      f"exported product graph ({rows} rows), oracle "
                      f"orc_hnsw_search, 1 thread (latency config)")
        del og
        cpu_baseline = {
            "value": round(nq_base / t_cpu, 3),
            "unit": "queries/s",
            "cores": cores,
            "kind": "port",
            "sample": sample,
        }
    elif world == 1 and not args.no_cpu_baseline:
        import oracle
        srows = min(args.cpu_sample_rows, rows)
        sample = oracle.gen_f32(args.seed, 0, srows, args.dim)
        q = queries[0]
        # warmup + timed queries; at the default workload srows == rows,
        # so the number is a FULL-corpus measurement (no extrapolation)
        oracle.topk_f32_mt(args.metric, sample, q, args.k)
        tcs = time.perf_counter()
        reps = 3
        used = 0
        for r in range(reps):
            _, _, used = oracle.topk_f32_mt(args.metric, sample,
                                            queries[r + 1], args.k)
        t_cpu_sample = (time.perf_counter() - tcs) / reps
        t_cpu_full = t_cpu_sample * (rows / srows)
        cpu_baseline = {
            "value": round(1.0 / t_cpu_full, 3),
            "unit": "queries/s",
            "cores": used,
            "kind": "port",
            "sample": f"{reps} queries x {srows} of {rows} rows "
                      f"({t_cpu_sample*1e3:.0f} ms/query on the sample; "
                      f"oracle orc_topk_f32_mt, OpenMP"
                      + ("" if srows == rows else "; extrapolated") + ")",
        }

    out = {
        "metric": "knn_qps",
        "value": round(qps, 3),
        "unit": "queries/s",
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(t_total / args.steps * 1e3, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,  # no published reference number
        "dtype": "f32",
        "data": "synthetic",
        "config": {
            "workload": (
                f"HNSW (M=16, ef={args.ef}) {args.metric} K={args.k}, "
                f"batch={args.batch} persistent-kernel path, {rows} "
                f"rows/GPU (configs[2] shape, batched)"
                if args.hnsw and args.batch > 0 else
                f"brute-force {args.metric} KNN, {rows} rows/GPU x "
                f"{args.dim}-dim f32, K={args.k}, batch={args.batch} "
                f"MFMA path (BASELINE configs[3])"
                if args.batch > 0 else
                f"HNSW (M=16, ef={args.ef}) {args.metric} K={args.k} single "
                f"query, {rows} rows/GPU (BASELINE configs[2] shape; "
                f"{'per-hop gather' if args.hnsw_perhop else 'persistent kernel'})"
                if args.hnsw else
                f"brute-force {args.metric} KNN, {rows} rows/GPU x "
                f"{args.dim}-dim f32, K={args.k}, single query "
                "(BASELINE configs[1])"),
            "batch": args.batch,
            "build": (("host" if args.host_build else "gpu")
                      + f"-snapshot chunk={args.chunk}") if args.hnsw
                     else None,
            "rows_total": rows * world,
            "rows_per_gpu": rows,
            "dim": args.dim,
            "k": args.k,
            "distance": args.metric,
            "parallelism": f"rowshard{world}" if world > 1 else "single",
            "stage_s": round(stage_s, 2),
        },
        "p50_ms": round(p50_ms, 3),
        "p95_ms": round(p95_ms, 3),
        "scan_kernel_ms_avg": round(scan_ms_avg, 3),
        "roofline": roofline,
        "cpu_baseline": cpu_baseline,
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
