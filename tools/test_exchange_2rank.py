#!/usr/bin/env python3
"""2-rank repartition-exchange check (run under torchrun on the GPU box):
orders sharded by orderkey are redistributed to custkey owners; verifies
ownership and row conservation."""
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, "/root/repo")


def main():
    dist.init_process_group("gloo")
    rank = dist.get_rank()
    torch.cuda.set_device(0)
    from opentenbase_amd import executor as ex
    from opentenbase_amd import fragment
    ex.init_device(0)
    od = ex.GpuOrders.generate(200000, 20000, rank=rank, nranks=2)
    keys, (dates,) = fragment.exchange_rows(od.t["o_custkey"],
                                            [od.t["o_orderdate"]])
    ok = bool(((keys % 2) == rank).all())
    tot = torch.tensor([len(keys)], dtype=torch.int64)
    dist.all_reduce(tot)
    assert len(keys) == len(dates)
    if rank == 0:
        print("exchange ok:", ok, "rows preserved:",
              int(tot.item()) == 200000, flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
