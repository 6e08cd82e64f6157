"""2-rank RCCL first-light probe on however many GPUs the box has.

Launched as:
  python -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
      --nproc-per-node 2 tools/rccl_probe2.py

On a 1-GPU box both ranks map cuda:0 (RCCL permits same-device ranks on
ROCm builds; if this box's build refuses, the probe reports that loudly and
exits 3 so the caller can distinguish "unsupported here" from "broken").
Exercises exactly the primitives the dist engines use: init_process_group
("nccl" = RCCL), broadcast, p2p isend/irecv pairs, all_reduce, barrier.
"""

import os
import sys

import torch
import torch.distributed as dist


def main() -> int:
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    n_dev = torch.cuda.device_count()
    dev = torch.device(f"cuda:{rank % n_dev}")
    torch.cuda.set_device(dev)
    try:
        dist.init_process_group("nccl", rank=rank, world_size=world)
    except Exception as e:  # noqa: BLE001
        print(f"[probe rank{rank}] init failed: {e}", flush=True)
        return 3
    try:
        # C1 analog: broadcast of a weight-sized vector
        w = (torch.arange(784, device=dev, dtype=torch.float32)
             if rank == 0 else torch.zeros(784, device=dev))
        dist.broadcast(w, src=0)
        assert float(w[783]) == 783.0, "broadcast payload wrong"
        # C3 analog: p2p gradient push worker->server
        if rank == 1:
            g = torch.full((784,), 2.0, device=dev)
            dist.send(g, dst=0)
        else:
            g = torch.zeros(784, device=dev)
            dist.recv(g, src=1)
            assert float(g.sum()) == 2.0 * 784, "p2p payload wrong"
        # C5 analog: allreduce
        s = torch.ones(1, device=dev) * (rank + 1)
        dist.all_reduce(s)
        assert float(s) == sum(range(1, world + 1)), "allreduce wrong"
        dist.barrier()
        torch.cuda.synchronize()
        print(f"[probe rank{rank}] RCCL ok: bcast+p2p+allreduce on {dev} "
              f"({n_dev} visible GPU(s), world={world})", flush=True)
        return 0
    except Exception as e:  # noqa: BLE001
        print(f"[probe rank{rank}] collective failed: {e}", flush=True)
        return 4
    finally:
        dist.destroy_process_group()


if __name__ == "__main__":
    sys.exit(main())
