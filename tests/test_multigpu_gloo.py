"""Multi-process sharded-search path on CPU: world_size=2 over gloo.

Covers the exact collective+merge sequence bench.py runs over RCCL on the
8-GPU box: per-rank local top-K -> all_gather of K (dist,id) pairs -> rank-0
merge with the reference tie-break order. Distances come from the oracle here
(CPU box); on GPU the same merge consumes sdbv_knn_bruteforce output.
"""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

import oracle
from surrealdb_amd.shard import merge_topk, shard_range

N, D, K, WORLD = 4000, 64, 10, 2


def _worker(rank, world, port, out_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        b, e = shard_range(N, rank, world)
        corpus = oracle.gen_f32(0x5DB1, b, e - b, D)
        q = oracle.gen_f32(0xBEEF, 0, 1, D)[0]
        ids, dists = oracle.topk_f32("cosine", corpus, q, K)
        ids = ids + b
        # pad to K (a shard can hold < K rows)
        pad = K - len(ids)
        if pad:
            ids = np.concatenate([ids, np.full(pad, np.iinfo(np.uint64).max,
                                               dtype=np.uint64)])
            dists = np.concatenate([dists, np.full(pad, np.inf)])
        local = torch.zeros(K, 2, dtype=torch.float64)
        local[:, 0] = torch.from_numpy(dists.copy())
        local[:, 1] = torch.from_numpy(ids.view(np.float64).copy())
        gathered = [torch.zeros_like(local) for _ in range(world)]
        dist.all_gather(gathered, local)
        if rank == 0:
            ids_list = [g[:, 1].numpy().view(np.uint64) for g in gathered]
            dists_list = [g[:, 0].numpy() for g in gathered]
            mids, mdists = merge_topk(ids_list, dists_list, K)
            out_q.put((mids, mdists))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_sharded_topk_equals_global():
    ctxm = mp.get_context("spawn")
    out_q = ctxm.Queue()
    port = 29511
    procs = [ctxm.Process(target=_worker, args=(r, WORLD, port, out_q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    mids, mdists = out_q.get(timeout=110)
    for p in procs:
        p.join(timeout=30)
    corpus = oracle.gen_f32(0x5DB1, 0, N, D)
    q = oracle.gen_f32(0xBEEF, 0, 1, D)[0]
    gids, gdists = oracle.topk_f32("cosine", corpus, q, K)
    assert np.array_equal(mids, gids)
    assert np.array_equal(mdists, gdists)


def _batch_worker(rank, world, port, out_q):
    """Batched-query variant: the exact per-query gather+merge sequence
    bench.py's one_batch runs over RCCL (b x K pairs per rank)."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        B = 6
        b0, e0 = shard_range(N, rank, world)
        corpus = oracle.gen_f32(0x5DB1, b0, e0 - b0, D)
        Q = oracle.gen_f32(0xBEEF, 0, B, D)
        flat = torch.zeros(B * K, 2, dtype=torch.float64)
        for j in range(B):
            ids, dists = oracle.topk_f32("cosine", corpus, Q[j], K)
            ids = ids + b0
            pad = K - len(ids)
            if pad:
                ids = np.concatenate(
                    [ids, np.full(pad, np.iinfo(np.uint64).max, np.uint64)])
                dists = np.concatenate([dists, np.full(pad, np.inf)])
            flat[j * K:(j + 1) * K, 0] = torch.from_numpy(dists.copy())
            flat[j * K:(j + 1) * K, 1] = torch.from_numpy(
                ids.view(np.float64).copy())
        gathered = [torch.zeros_like(flat) for _ in range(world)]
        dist.all_gather(gathered, flat)
        if rank == 0:
            out = []
            g = [t.numpy().reshape(B, K, 2) for t in gathered]
            for j in range(B):
                out.append(merge_topk(
                    [x[j, :, 1].copy().view(np.uint64) for x in g],
                    [x[j, :, 0] for x in g], K))
            out_q.put(out)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_sharded_batch_topk_equals_global():
    ctxm = mp.get_context("spawn")
    out_q = ctxm.Queue()
    port = 29513
    procs = [ctxm.Process(target=_batch_worker, args=(r, WORLD, port, out_q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    merged = out_q.get(timeout=110)
    for p in procs:
        p.join(timeout=30)
    corpus = oracle.gen_f32(0x5DB1, 0, N, D)
    Q = oracle.gen_f32(0xBEEF, 0, 6, D)
    for j in range(6):
        gids, gdists = oracle.topk_f32("cosine", corpus, Q[j], K)
        assert np.array_equal(merged[j][0], gids), f"q{j}"
        assert np.array_equal(merged[j][1], gdists), f"q{j}"


@pytest.mark.timeout(180)
def test_sharded_topk_world4_with_duplicates():
    """world_size=4 (half the 8-GPU node's rank count) with exact
    duplicate rows straddling shard boundaries: the cross-rank merge must
    keep the reference (dist total_cmp asc, id asc) order for ties that
    live on DIFFERENT ranks."""
    ctxm = mp.get_context("spawn")
    out_q = ctxm.Queue()
    port = 29517
    world = 4
    procs = [ctxm.Process(target=_dup_worker, args=(r, world, port, out_q))
             for r in range(world)]
    for p in procs:
        p.start()
    mids, mdists = out_q.get(timeout=170)
    for p in procs:
        p.join(timeout=30)
    corpus = _dup_corpus()
    q = corpus[7].copy()
    gids, gdists = oracle.topk_f32("cosine", corpus, q, K)
    assert np.array_equal(mids, gids)
    assert np.array_equal(mdists, gdists)
    # the duplicate of row 7 lives on another shard; both must surface as
    # the equal-distance leaders, ascending by global id (cosine
    # self-distance under the restated f32 chain is ~1e-7, not exactly 0)
    assert mids[0] == 7 and mids[1] == 3007
    assert mdists[0] == mdists[1]


def _dup_corpus():
    base = oracle.gen_f32(0x5DB1, 0, N, D)
    base[3007] = base[7]  # exact duplicate across shard boundary (w=4)
    return base


def _dup_worker(rank, world, port, out_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        full = _dup_corpus()
        b, e = shard_range(N, rank, world)
        corpus = full[b:e]
        q = full[7].copy()
        ids, dists = oracle.topk_f32("cosine", corpus, q, K)
        ids = ids + b
        pad = K - len(ids)
        if pad:
            ids = np.concatenate([ids, np.full(pad, np.iinfo(np.uint64).max,
                                               dtype=np.uint64)])
            dists = np.concatenate([dists, np.full(pad, np.inf)])
        local = torch.zeros(K, 2, dtype=torch.float64)
        local[:, 0] = torch.from_numpy(dists.copy())
        local[:, 1] = torch.from_numpy(ids.view(np.float64).copy())
        gathered = [torch.zeros_like(local) for _ in range(world)]
        dist.all_gather(gathered, local)
        if rank == 0:
            ids_list = [g[:, 1].numpy().view(np.uint64) for g in gathered]
            dists_list = [g[:, 0].numpy() for g in gathered]
            out_q.put(merge_topk(ids_list, dists_list, K))
    finally:
        dist.destroy_process_group()
