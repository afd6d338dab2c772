"""Coordinator shard-merge over RCCL/xGMI — the MI355X-native replacement of
the reference's RemoteSubplan recv + FN transport (execFragment.c:3877,
forward/*; SURVEY §2 'Fragment exchange'). Semantics: per-shard partial-agg
streams are CONCATENATED at the CN and combined by the Finalize Aggregate
(float8pl / int8pl / float8_combine — nodeAgg.c:3912,4260).

One process per GPU (1 DN shard ↔ 1 GPU); collectives via torch.distributed —
backend "nccl" IS RCCL on ROCm (xGMI inside the node); CPU tests use "gloo"
with the same code path.
"""
import torch
import torch.distributed as dist

from .executor import q1_finalize, q1_rows_from_state


def is_dist():
    return dist.is_available() and dist.is_initialized()


def merge_q1_partials(sums, counts):
    """All-gather the dense per-rank Q1 partial states (6×5 sums + 6 counts —
    the 'Remote Subquery Scan' payload: 4 rows/DN) and combine on every rank
    (combine is cheap and keeping it symmetric avoids a broadcast back).

    Dense slot layout makes the combine a pure elementwise sum: float8pl for
    sums, int8pl for counts, slot-aligned groups."""
    if not is_dist() or dist.get_world_size() == 1:
        return q1_rows_from_state(sums, counts)
    if dist.get_backend() == "gloo" and sums.is_cuda:
        sums, counts = sums.cpu(), counts.cpu()
    world = dist.get_world_size()
    gs = [torch.empty_like(sums) for _ in range(world)]
    gc = [torch.empty_like(counts) for _ in range(world)]
    dist.all_gather(gs, sums)
    dist.all_gather(gc, counts)
    tot_s = torch.zeros_like(sums)
    tot_c = torch.zeros_like(counts)
    for s in gs:
        tot_s += s            # float8pl chain over shards
    for c in gc:
        tot_c += c            # int8pl
    return q1_rows_from_state(tot_s, tot_c)


def merge_q9_partials(sums, counts):
    """Q9-mix shard merge: all-gather + elementwise combine of the dense
    [7]-year partial states (same RemoteSubplan payload shape as Q1)."""
    from .executor import q9_rows_from_state
    if not is_dist() or dist.get_world_size() == 1:
        return q9_rows_from_state(sums, counts)
    if dist.get_backend() == "gloo" and sums.is_cuda:
        sums, counts = sums.cpu(), counts.cpu()
    dist.all_reduce(sums)      # float8pl over shards
    dist.all_reduce(counts)    # int8pl
    return q9_rows_from_state(sums, counts)


def allgather_variable(t):
    """All-gather a 1-D tensor with per-rank variable length (the FN-page
    concatenation semantics). Returns the concatenation over ranks."""
    if not is_dist() or dist.get_world_size() == 1:
        return t
    if dist.get_backend() == "gloo" and t.is_cuda:
        t = t.cpu()
    world = dist.get_world_size()
    n = torch.tensor([t.numel()], dtype=torch.int64, device=t.device)
    ns = [torch.empty_like(n) for _ in range(world)]
    dist.all_gather(ns, n)
    counts = [int(x.item()) for x in ns]
    mx = max(counts) if counts else 0
    pad = torch.empty(mx, dtype=t.dtype, device=t.device)
    pad[: t.numel()] = t
    outs = [torch.empty_like(pad) for _ in range(world)]
    dist.all_gather(outs, pad)
    return torch.cat([o[:c] for o, c in zip(outs, counts)])


def broadcast_customer_keys(local_keys):
    """Q3 replicated build side: all-gather each rank's filtered customer
    keys (reference analog: replicated distribution / 'Distribute results by'
    exchange, xl_join.out:13-20). Shards are disjoint (custkey % nranks), so
    concatenation = the full filtered key set."""
    return allgather_variable(local_keys)


def merge_q3_topk(candidates, k=10):
    """All-gather per-rank top-k candidate rows (as raw bytes) and take the
    global top-k (merge-sorted recv analog, execFragment.c:4035-4059)."""
    import numpy as np
    dt = np.dtype([("l_orderkey", "i8"), ("revenue", "f8"),
                   ("o_orderdate", "i4"), ("o_shippriority", "i4")])
    if is_dist() and dist.get_world_size() > 1:
        raw = torch.from_numpy(
            np.ascontiguousarray(candidates).view(np.uint8).reshape(-1).copy())
        dev = "cuda" if torch.cuda.is_available() and \
            dist.get_backend() == "nccl" else "cpu"
        raw = raw.to(dev)
        allraw = allgather_variable(raw).cpu().numpy().tobytes()
        cands = np.frombuffer(allraw, dtype=dt).copy()
    else:
        cands = np.asarray(candidates, dtype=dt)
    order = np.lexsort((cands["l_orderkey"], cands["o_orderdate"],
                        -cands["revenue"]))
    return cands[order][:k]


def finalize_q1(rows):
    return q1_finalize(rows)


def all_to_all_variable(t, send_counts):
    """All-to-all with per-rank variable segment lengths: rank r sends
    t[offsets[r]:offsets[r]+send_counts[r]] to rank r and receives the
    concatenation of every rank's segment addressed to it. The RCCL
    all-to-allv of the repartition exchange (FN-transport
    semantics of forward/*). nccl: dist.all_to_all_single with splits;
    gloo (CPU tests): emulated via all-gather + slicing (gloo has no
    all-to-all)."""
    if not is_dist() or dist.get_world_size() == 1:
        return t
    world = dist.get_world_size()
    assert len(send_counts) == world
    if dist.get_backend() == "nccl":
        sc = torch.tensor(send_counts, dtype=torch.int64, device=t.device)
        rc = torch.empty_like(sc)
        dist.all_to_all_single(rc, sc)
        recv_counts = [int(x) for x in rc.cpu()]
        out = torch.empty(sum(recv_counts), dtype=t.dtype, device=t.device)
        dist.all_to_all_single(out, t, output_split_sizes=recv_counts,
                               input_split_sizes=list(send_counts))
        return out
    # gloo emulation: gather everything + counts, slice my segments
    if t.is_cuda:
        t = t.cpu()
    rank = dist.get_rank()
    cnts = torch.tensor(send_counts, dtype=torch.int64)
    all_cnts = [torch.empty_like(cnts) for _ in range(world)]
    dist.all_gather(all_cnts, cnts)
    alldata = allgather_variable(t)
    out = []
    pos = 0
    for src in range(world):
        c = [int(x) for x in all_cnts[src]]
        seg_start = pos + sum(c[:rank])
        out.append(alldata[seg_start: seg_start + c[rank]])
        pos += sum(c)
    return torch.cat(out) if out else t[:0]


def exchange_rows(keys, payload_tensors):
    """Full repartition exchange: GPU partition by key % world, native
    gathers, all-to-allv of every column. Returns (keys', payloads') — the
    rows this rank owns after redistribution."""
    from . import executor as ex
    perm, counts = ex.partition_by_key(keys)
    out_keys = all_to_all_variable(ex.gather(keys, perm), counts)
    out_payloads = [all_to_all_variable(ex.gather(p, perm), counts)
                    for p in payload_tensors]
    return out_keys, out_payloads
