"""torchrun-launched probe of the distributed bench mechanics on CPU: reads
RANK/WORLD_SIZE/MASTER_* from the env exactly like bench.py, initializes the
gloo group, and runs the sharded top-k all-gather merge path
(matrixone_amd.distributed) end-to-end. Launched by
test_distributed_cpu.py::test_torchrun_launch — the same
`python -m torch.distributed.run --nnodes=1 --nproc-per-node N
--master-addr 127.0.0.1` invocation the driver uses for bench.py --gpus N.
"""
import os
import sys

import numpy as np
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    world = int(os.environ["WORLD_SIZE"])
    rank = int(os.environ["RANK"])
    assert "MASTER_ADDR" in os.environ and "MASTER_PORT" in os.environ
    dist.init_process_group("gloo")
    from matrixone_amd.distributed import allgather_merge

    k, nq = 5, 8
    rng = np.random.Generator(np.random.PCG64(7 + rank))
    ids = rng.integers(0, 1000, (nq, k)).astype(np.int64) + rank * 10000
    dists = np.sort(rng.random((nq, k)).astype(np.float32), axis=1) + rank
    m_ids, m_dists = allgather_merge(ids, dists, k)
    assert m_ids.shape == (nq, k)
    # rank 0's first-column entries are the global minima (dists offset by
    # rank, so rank 0 wins every slot)
    assert (m_ids < 10000).all(), "merge must favour the rank-0 shard (offset dists)"
    assert (np.diff(m_dists, axis=1) >= 0).all()
    dist.barrier()
    if rank == 0:
        print("TORCHRUN_PROBE_OK", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
