"""Multi-process CPU tests of the Coordinator shard-merge (fragment.py) over
the gloo backend, world_size 2 — covers the N>1 distributed path that the
driver's round-end scaling bench exercises over RCCL."""
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)


def _worker_q1(rank, world, port, q):
    from opentenbase_amd import fragment
    _init(rank, world, port)
    # rank-local partial states, dense slot layout
    sums = torch.zeros((6, 5), dtype=torch.float64)
    counts = torch.zeros(6, dtype=torch.int64)
    sums[0, 0] = 1.5 * (rank + 1)
    counts[0] = 10 * (rank + 1)
    sums[3, 2] = 2.0
    counts[3] = rank  # only rank 1 populates slot 3
    rows = fragment.merge_q1_partials(sums, counts)
    if rank == 0:
        q.put(rows)
    torch.distributed.destroy_process_group()


def _worker_allgatherv(rank, world, port, q):
    from opentenbase_amd import fragment
    _init(rank, world, port)
    t = torch.arange(3 + 2 * rank, dtype=torch.int64) + 100 * rank
    out = fragment.allgather_variable(t)
    if rank == 0:
        q.put(out.numpy())
    torch.distributed.destroy_process_group()


def _worker_topk(rank, world, port, q):
    from opentenbase_amd import fragment
    dt = np.dtype([("l_orderkey", "i8"), ("revenue", "f8"),
                   ("o_orderdate", "i4"), ("o_shippriority", "i4")])
    _init(rank, world, port)
    cands = np.zeros(3, dtype=dt)
    cands["l_orderkey"] = np.arange(3) + 10 * rank
    cands["revenue"] = [5.0 + rank, 1.0, 3.0 + 2 * rank]
    top = fragment.merge_q3_topk(cands, k=4)
    if rank == 0:
        q.put(top)
    torch.distributed.destroy_process_group()


def _run(worker, world=2):
    # retry with fresh ports: a lingering TIME_WAIT on a reused MASTER_PORT
    # can fail the rendezvous (rare flake under suite load)
    last = None
    for attempt in range(3):
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        port = 29511 + np.random.randint(0, 2000)
        procs = [ctx.Process(target=worker, args=(r, world, port, q))
                 for r in range(world)]
        for p in procs:
            p.start()
        try:
            res = q.get(timeout=120)
        except Exception as e:
            last = e
            for p in procs:
                p.terminate()
                p.join(timeout=30)
            continue
        ok = True
        for p in procs:
            p.join(timeout=60)
            ok = ok and p.exitcode == 0
        if ok:
            return res
        last = AssertionError([p.exitcode for p in procs])
    raise last


@pytest.mark.timeout(180)
def test_merge_q1_partials_two_ranks():
    rows = _run(_worker_q1)
    # slot 0 = (A,F): counts 10+20, sums 1.5+3.0; slot 3 = (N,O): count 0+1
    af = [r for r in rows if r["l_returnflag"] == "A"][0]
    assert af["count_order"] == 30
    assert af["sum_qty"] == 4.5
    no = [r for r in rows if r["l_returnflag"] == "N"][0]
    assert no["count_order"] == 1
    assert no["sum_disc_price"] == 4.0  # 2.0 from each rank's tensor


@pytest.mark.timeout(180)
def test_allgather_variable():
    out = _run(_worker_allgatherv)
    assert out.tolist() == [0, 1, 2, 100, 101, 102, 103, 104]


@pytest.mark.timeout(180)
def test_merge_q3_topk():
    top = _run(_worker_topk)
    assert top["revenue"].tolist() == [6.0, 5.0, 5.0, 3.0]


def _worker_q9(rank, world, port, q):
    from opentenbase_amd import fragment
    _init(rank, world, port)
    sums = torch.zeros(7, dtype=torch.float64)
    counts = torch.zeros(7, dtype=torch.int64)
    sums[0] = 1.25 * (rank + 1)
    counts[0] = 7 * (rank + 1)
    sums[6] = 3.0
    counts[6] = rank  # only rank 1 populates 1998
    rows = fragment.merge_q9_partials(sums, counts)
    if rank == 0:
        q.put(rows)
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(180)
def test_merge_q9_partials_two_ranks():
    rows = _run(_worker_q9)
    y92 = [r for r in rows if r["o_year"] == 1992][0]
    assert y92["count_rows"] == 21
    assert y92["sum_revenue"] == 3.75
    y98 = [r for r in rows if r["o_year"] == 1998][0]
    assert y98["count_rows"] == 1
    assert y98["sum_revenue"] == 6.0
    assert len(rows) == 2


def _worker_a2a(rank, world, port, q):
    from opentenbase_amd import fragment
    _init(rank, world, port)
    # rank r sends [r*100 + 0..2] to rank 0 and [r*100 + 10..12] to rank 1
    t = torch.tensor([rank * 100 + i for i in (0, 1, 2)] +
                     [rank * 100 + i for i in (10, 11, 12)], dtype=torch.int64)
    out = fragment.all_to_all_variable(t, [3, 3])
    if rank == 0:
        q.put(out.numpy())
    torch.distributed.destroy_process_group()


def _run_n(worker, world):
    return _run(worker, world=world)


def _worker_q1_w4(rank, world, port, q):
    from opentenbase_amd import fragment
    _init(rank, world, port)
    sums = torch.zeros((6, 5), dtype=torch.float64)
    counts = torch.zeros(6, dtype=torch.int64)
    sums[0, 0] = float(rank + 1)
    counts[0] = rank + 1
    rows = fragment.merge_q1_partials(sums, counts)
    if rank == 0:
        q.put(rows)
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(240)
def test_merge_q1_partials_four_ranks():
    """World-size 4 (the driver's N=4 scale point): elementwise combine
    over four shards."""
    rows = _run_n(_worker_q1_w4, 4)
    af = [r for r in rows if r["l_returnflag"] == "A"][0]
    assert af["count_order"] == 1 + 2 + 3 + 4
    assert af["sum_qty"] == 1.0 + 2.0 + 3.0 + 4.0


def _worker_a2a_w4(rank, world, port, q):
    from opentenbase_amd import fragment
    _init(rank, world, port)
    # rank r sends one element tagged (r, dest) to every dest
    t = torch.tensor([rank * 10 + d for d in range(world)], dtype=torch.int64)
    out = fragment.all_to_all_variable(t, [1] * world)
    if rank == 1:
        q.put(out.numpy())
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(240)
def test_all_to_all_variable_four_ranks():
    out = _run_n(_worker_a2a_w4, 4)
    # rank 1 receives element (r*10 + 1) from every rank r, in rank order
    assert out.tolist() == [1, 11, 21, 31]


@pytest.mark.timeout(180)
def test_all_to_all_variable_gloo():
    out = _run(_worker_a2a)
    # rank 0 receives rank 0's first segment then rank 1's first segment
    assert out.tolist() == [0, 1, 2, 100, 101, 102]


def _worker_q1_w8(rank, world, port, q):
    from opentenbase_amd import fragment
    _init(rank, world, port)
    sums = torch.zeros((6, 5), dtype=torch.float64)
    counts = torch.zeros(6, dtype=torch.int64)
    sums[0, 0] = float(rank + 1)
    counts[0] = rank + 1
    rows = fragment.merge_q1_partials(sums, counts)
    if rank == 0:
        q.put(rows)
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(300)
def test_merge_q1_partials_eight_ranks():
    """World-size 8 — the driver's full-node scale point (8 DataNode
    shards / 8 GPUs): the Q1 finalize combine over eight shards."""
    rows = _run_n(_worker_q1_w8, 8)
    af = [r for r in rows if r["l_returnflag"] == "A"][0]
    assert af["count_order"] == sum(range(1, 9))
    assert af["sum_qty"] == float(sum(range(1, 9)))


def _worker_a2a_w8(rank, world, port, q):
    from opentenbase_amd import fragment
    _init(rank, world, port)
    t = torch.tensor([rank * 10 + d for d in range(world)], dtype=torch.int64)
    out = fragment.all_to_all_variable(t, [1] * world)
    if rank == 3:
        q.put(out.numpy())
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(300)
def test_all_to_all_variable_eight_ranks():
    """World-size 8 repartition exchange (the §8f.1 path at the full-node
    width): rank 3 receives the (r, 3)-tagged element from every rank in
    rank order."""
    out = _run_n(_worker_a2a_w8, 8)
    assert out.tolist() == [3, 13, 23, 33, 43, 53, 63, 73]


def _worker_a2a_zero(rank, world, port, q):
    from opentenbase_amd import fragment
    _init(rank, world, port)
    # skewed repartition edge: rank 1 has NO rows at all; rank 0 sends
    # nothing to itself and 3 rows to rank 1
    if rank == 0:
        t = torch.tensor([10, 11, 12], dtype=torch.int64)
        out = fragment.all_to_all_variable(t, [0, 3])
    else:
        t = torch.empty(0, dtype=torch.int64)
        out = fragment.all_to_all_variable(t, [0, 0])
    q.put((rank, out.numpy()))
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(180)
def test_all_to_all_zero_length_segments():
    """Zero-length splits and a completely empty rank (the skewed
    distribution-key case of BASELINE config 5 at its extreme) must
    round-trip: rank 0 ends with nothing, rank 1 with rank 0's rows."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29911 + np.random.randint(400, 800)
    procs = [ctx.Process(target=_worker_a2a_zero, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    res = dict(q.get(timeout=180) for _ in range(2))
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert res[0].tolist() == []
    assert res[1].tolist() == [10, 11, 12]


def _worker_topk_empty(rank, world, port, q):
    from opentenbase_amd import fragment
    dt = np.dtype([("l_orderkey", "i8"), ("revenue", "f8"),
                   ("o_orderdate", "i4"), ("o_shippriority", "i4")])
    _init(rank, world, port)
    # rank 1's shard has NO qualifying candidates (selective segment on a
    # small shard) — the merge must still return rank 0's rows
    n = 3 if rank == 0 else 0
    cands = np.zeros(n, dtype=dt)
    if n:
        cands["l_orderkey"] = [1, 2, 3]
        cands["revenue"] = [5.0, 9.0, 7.0]
    top = fragment.merge_q3_topk(cands, k=2)
    if rank == 0:
        q.put(top)
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(180)
def test_merge_q3_topk_empty_shard():
    top = _run(_worker_topk_empty)
    assert top["revenue"].tolist() == [9.0, 7.0]
    assert top["l_orderkey"].tolist() == [2, 3]


def _worker_dist_leftjoin(rank, world, port, q):
    """Repartitioned LEFT JOIN across ranks: both sides exchanged by
    owner = key % world (NULL keys to rank 0 — the reference routes NULL
    distribution keys to a designated node), rank-local ora_join_ext(left),
    union of pair sets == the full-table oracle left join. Models the
    'Distribute results by H' + outer-join plan shape (xl_join.out:13-20,
    nodeHashjoin.c HJ_FILL_OUTER_TUPLE)."""
    from opentenbase_amd import fragment
    from oracle import oracle_py as ora
    _init(rank, world, port)
    rng = np.random.default_rng(5)
    nb, npr = 300, 900
    bk = rng.integers(0, 40, nb)
    pk = rng.integers(0, 40, npr)
    bn = (rng.random(nb) < 0.1).astype(np.uint8)
    pn = (rng.random(npr) < 0.1).astype(np.uint8)

    def shard_and_exchange(keys, nulls):
        # this rank starts with the round-robin shard, then exchanges by
        # owner = key % world (NULL rows -> rank 0)
        mine = np.arange(len(keys)) % world == rank
        ids = np.nonzero(mine)[0]
        owner = np.where(nulls[ids] == 1, 0, keys[ids] % world)
        order = np.argsort(owner, kind="stable")
        ids = ids[order]
        counts = [int((owner == r).sum()) for r in range(world)]
        out_k = fragment.all_to_all_variable(
            torch.as_tensor(keys[ids]), counts)
        out_i = fragment.all_to_all_variable(
            torch.as_tensor(ids), counts)
        out_n = fragment.all_to_all_variable(
            torch.as_tensor(nulls[ids]), counts)
        return (out_k.numpy(), out_i.numpy(),
                out_n.numpy().astype(np.uint8))

    lbk, lbi, lbn = shard_and_exchange(bk, bn)
    lpk, lpi, lpn = shard_and_exchange(pk, pn)
    bi, pi = ora.join_ext(lbk, lpk, 1, bnull=lbn, pnull=lpn)  # left
    # map local indices back to GLOBAL row ids
    pairs = [(int(lbi[b]) if b >= 0 else -1, int(lpi[p])) for b, p in
             zip(bi.tolist(), pi.tolist())]
    gathered = fragment.allgather_variable(
        torch.as_tensor(np.array(pairs, dtype=np.int64).reshape(-1)))
    if rank == 0:
        allp = gathered.numpy().reshape(-1, 2)
        got = sorted(map(tuple, allp.tolist()))
        obi, opi = ora.join_ext(bk, pk, 1, bnull=bn, pnull=pn)
        exp = sorted(zip(obi.tolist(), opi.tolist()))
        q.put((got, exp))
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(180)
def test_distributed_left_join_two_ranks():
    got, exp = _run(_worker_dist_leftjoin)
    assert got == exp


def _worker_dec_combine(rank, world, port, q):
    """Decimal partial states combine EXACTLY across ranks: each rank
    aggregates its shard with the int128 oracle, states are all-gathered
    (as the CN merge ships them) and summed with unbounded ints; result
    must equal the whole-table aggregate bit-exactly (the numeric
    combine-phase analog: int128 state addition cannot lose precision)."""
    from opentenbase_amd import fragment
    from oracle import oracle_py as ora
    _init(rank, world, port)
    rng = np.random.default_rng(7)
    n = 4000
    keys = rng.integers(0, 6, n)
    vals = rng.integers(-2**62, 2**62, n)     # sums leave int64 quickly
    mine = np.arange(n) % world == rank
    local = ora.agg_i64_dec(keys[mine], vals[mine])
    # ship (key, count_star, count_v, sum_hi, sum_lo) rows
    rows = np.array([(g.key, g.count_star, g.count_v, g.sum_hi,
                      np.int64(g.sum_lo - 2**63)) for g in local],
                    dtype=np.int64).reshape(-1)
    gathered = fragment.allgather_variable(torch.as_tensor(rows))
    if rank == 0:
        allrows = gathered.numpy().reshape(-1, 5)
        comb = {}
        for k, cs, cv, hi, lo_biased in allrows.tolist():
            s128 = (int(hi) << 64) | (int(lo_biased) + 2**63)
            c = comb.setdefault(int(k), [0, 0, 0])
            c[0] += int(cs)
            c[1] += int(cv)
            c[2] += s128
        full = ora.agg_i64_dec(keys, vals)
        ok = len(full) == len(comb)
        for g in full:
            cs, cv, s = comb[int(g.key)]
            ok = ok and cs == g.count_star and cv == g.count_v \
                and s == g.sum128
        q.put(ok)
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(180)
def test_distributed_dec_combine_exact():
    assert _run(_worker_dec_combine)
