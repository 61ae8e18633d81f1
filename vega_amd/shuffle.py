"""Rank-per-GPU shuffle exchange (replaces the reference's HTTP-pull shuffle
plane: shuffle_manager.rs:86-118 server + shuffle_fetcher.rs:61-90 client +
map_output_tracker.rs URI registry).

MI355X-native plan: one process per GPU over
torch.distributed; bucket b of every map shard is owned by rank b; the
exchange is ONE counts all-to-all (replacing the MapOutputTracker) followed
by one all-to-all-v of the bucket payloads (RCCL over xGMI on GPUs; gloo on
CPU for tests — same code path).
"""
import numpy as np
import torch
import torch.distributed as dist


def hash_u64_np(keys):
    """vectorized splitmix64 (bit-identical to vega_common.h vega_hash_u64)"""
    x = np.asarray(keys).astype(np.int64).view(np.uint64).copy()
    with np.errstate(over="ignore"):
        x = x + np.uint64(0x9E3779B97F4A7C15)
        x = (x ^ (x >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
        x = (x ^ (x >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
        x = x ^ (x >> np.uint64(31))
    return x


def bucket_of_np(keys, nparts):
    return (hash_u64_np(keys) % np.uint64(nparts)).astype(np.int64)


def partition_cpu(k, v, nparts):
    """CPU reference partition (tests / gloo path): bucket-contiguous reorder
    + per-bucket counts. Stable within buckets (row order preserved), matching
    the GPU k_scatter's stable counting scatter."""
    k = np.asarray(k, dtype=np.int64)
    v = np.asarray(v, dtype=np.int64)
    b = bucket_of_np(k, nparts)
    order = np.argsort(b, kind="stable")
    counts = np.bincount(b, minlength=nparts).astype(np.int64)
    return k[order], v[order], counts


def choose_splitters(keys, world, group=None, samples=4096):
    """Range-partitioner boundaries for sort_by_key's exchange (the analogue
    of Spark's RangePartitioner sampling; reference has only take_ordered,
    rdd.rs:1106-1153): every rank contributes up to `samples` strided key
    samples plus its row count; quantiles are taken over the VALID samples
    only, each weighted by the rows it represents (n_r / s_r), so short or
    empty ranks neither pad nor skew the boundaries. Returns an int64 tensor
    of world-1 ascending splitters on keys.device."""
    n = keys.numel()
    dev = keys.device
    s_local = min(samples, n)
    if s_local > 0:
        idx = torch.linspace(0, n - 1, steps=s_local, dtype=torch.int64, device=dev)
        local = keys[idx]
        if s_local < samples:  # pad to fixed size for all_gather (dropped below)
            local = torch.cat([local, local.new_zeros(samples - s_local)])
    else:
        local = torch.zeros(samples, dtype=torch.int64, device=dev)
    meta = torch.tensor([n], dtype=torch.int64, device=dev)
    gathered = [torch.empty_like(local) for _ in range(world)]
    gmeta = [torch.empty_like(meta) for _ in range(world)]
    dist.all_gather(gathered, local, group=group)
    dist.all_gather(gmeta, meta, group=group)
    vals, wts = [], []
    for r in range(world):
        n_r = int(gmeta[r].item())
        s_r = min(samples, n_r)
        if s_r == 0:
            continue
        vals.append(gathered[r][:s_r])
        wts.append(torch.full((s_r,), n_r / s_r, dtype=torch.float64, device=dev))
    if not vals:
        return torch.zeros(world - 1, dtype=torch.int64, device=dev)
    allk = torch.cat(vals)
    allw = torch.cat(wts)
    order = allk.argsort()
    allk = allk[order]
    cumw = allw[order].cumsum(0)
    total = float(cumw[-1].item())
    targets = torch.tensor([i * total / world for i in range(1, world)],
                           dtype=torch.float64, device=dev)
    pos = torch.searchsorted(cumw, targets).clamp_(0, allk.numel() - 1)
    return allk[pos].contiguous()


def partition_range_cpu(k, v, splitters):
    """CPU reference of the range partition (tests / gloo path): bucket =
    #splitters <= key, stable reorder + counts (matches RangeDigit)."""
    k = np.asarray(k, dtype=np.int64)
    v = np.asarray(v, dtype=np.int64)
    spl = np.asarray(splitters, dtype=np.int64)
    b = np.searchsorted(spl, k, side="right")
    order = np.argsort(b, kind="stable")
    counts = np.bincount(b, minlength=len(spl) + 1).astype(np.int64)
    return k[order], v[order], counts


def all_to_all_kv(send_k, send_v, send_counts, group=None):
    """Exchange bucket-contiguous (k, v) rows: counts all-to-all (replacing
    the MapOutputTracker), then the payload exchange as grouped P2P
    send/recv (RCCL over xGMI on CUDA, gloo on CPU — same code path).

    The rank's OWN bucket never touches the collective: it moves with a
    device copy (all_to_all_single was measured moving the self-shard at
    ~8 GB/s — on the 8-GPU node 1/8 of every exchange would have crawled
    while the 7 xGMI links idled)."""
    dev = send_k.device
    world = dist.get_world_size(group=group)
    rank = dist.get_rank(group=group)
    sc = torch.as_tensor(send_counts, dtype=torch.int64, device=dev)
    rc = torch.empty_like(sc)
    dist.all_to_all_single(rc, sc, group=group)
    in_splits = [int(x) for x in sc.tolist()]
    out_splits = [int(x) for x in rc.tolist()]
    nrecv = sum(out_splits)
    recv_k = torch.empty(nrecv, dtype=send_k.dtype, device=dev)
    recv_v = torch.empty(nrecv, dtype=send_v.dtype, device=dev)
    soff = [0] * world
    roff = [0] * world
    for p in range(1, world):
        soff[p] = soff[p - 1] + in_splits[p - 1]
        roff[p] = roff[p - 1] + out_splits[p - 1]
    ops = []
    for p in range(world):
        if p == rank:
            continue
        if in_splits[p]:
            ops.append(dist.P2POp(dist.isend, send_k[soff[p]:soff[p] + in_splits[p]],
                                  p, group=group))
            ops.append(dist.P2POp(dist.isend, send_v[soff[p]:soff[p] + in_splits[p]],
                                  p, group=group))
        if out_splits[p]:
            ops.append(dist.P2POp(dist.irecv, recv_k[roff[p]:roff[p] + out_splits[p]],
                                  p, group=group))
            ops.append(dist.P2POp(dist.irecv, recv_v[roff[p]:roff[p] + out_splits[p]],
                                  p, group=group))
    if in_splits[rank]:  # self bucket: plain device copy, off the wire
        recv_k[roff[rank]:roff[rank] + in_splits[rank]].copy_(
            send_k[soff[rank]:soff[rank] + in_splits[rank]])
        recv_v[roff[rank]:roff[rank] + in_splits[rank]].copy_(
            send_v[soff[rank]:soff[rank] + in_splits[rank]])
    if ops:
        for req in dist.batch_isend_irecv(ops):
            req.wait()
    return recv_k, recv_v
