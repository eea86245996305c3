"""Process-group runtime: init, rank helpers, collectives.

Reference parity: env-var launch (swin main.py:25-27,317-320),
SLURM detect (RetinaNet train_utils/distributed_utils.py:300-323), rank helpers
(fasterRcnn utils/distributed_utils.py:279-296), reduce_value
(others/train_with_DDP utils/distributed_utils.py:72-77), object all-gather
(RetinaNet :73-113, YOLOX yolox/utils/dist.py:139-226),
torch_distributed_zero_first (swin utils/torch_utils.py:16-23).

MI355X design: backend "nccl" IS RCCL on ROCm; one process per GPU over xGMI.
CPU/test path uses gloo. MASTER_ADDR defaults to 127.0.0.1 (container hostname
may not resolve).
"""
from __future__ import annotations

import os
import pickle
from contextlib import contextmanager

import torch
import torch.distributed as dist


def is_dist() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_dist() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_dist() else 1


def get_local_rank() -> int:
    if not is_dist():
        return 0
    return int(os.environ.get("LOCAL_RANK", get_rank() % max(1, torch.cuda.device_count() or 1)))


def is_main_process() -> bool:
    return get_rank() == 0


def barrier() -> None:
    if is_dist():
        dist.barrier()


def init_distributed(backend: str | None = None) -> dict:
    """Initialise from torchrun env vars (RANK/LOCAL_RANK/WORLD_SIZE) or SLURM.

    Returns {'rank','local_rank','world_size','distributed'}. Safe to call in a
    single-process run: returns distributed=False without creating a group.
    """
    if "RANK" in os.environ and "WORLD_SIZE" in os.environ:
        rank = int(os.environ["RANK"])
        world_size = int(os.environ["WORLD_SIZE"])
        local_rank = int(os.environ.get("LOCAL_RANK", 0))
    elif "SLURM_PROCID" in os.environ:
        rank = int(os.environ["SLURM_PROCID"])
        world_size = int(os.environ.get("SLURM_NTASKS", 1))
        local_rank = rank % max(1, torch.cuda.device_count() or 1)
    else:
        return {"rank": 0, "local_rank": 0, "world_size": 1, "distributed": False}
    if world_size <= 1:
        return {"rank": 0, "local_rank": 0, "world_size": 1, "distributed": False}

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    dist.init_process_group(backend=backend, init_method="env://",
                            rank=rank, world_size=world_size)
    dist.barrier()
    return {"rank": rank, "local_rank": local_rank, "world_size": world_size,
            "distributed": True}


def cleanup() -> None:
    if is_dist():
        dist.destroy_process_group()


def reduce_value(value: torch.Tensor, average: bool = True) -> torch.Tensor:
    """All-reduce a tensor across ranks (SUM, optionally averaged)."""
    if not is_dist():
        return value
    with torch.no_grad():
        dist.all_reduce(value)
        if average:
            value = value / get_world_size()
    return value


def reduce_dict(d: dict, average: bool = True) -> dict:
    """All-reduce a dict of scalar tensors with ONE collective (stacked),
    reference RetinaNet train_utils/distributed_utils.py:116-141."""
    if not is_dist() or len(d) == 0:
        return d
    with torch.no_grad():
        keys = sorted(d.keys())
        vals = torch.stack([d[k].detach().float() for k in keys])
        dist.all_reduce(vals)
        if average:
            vals = vals / get_world_size()
        return {k: vals[i] for i, k in enumerate(keys)}


def all_gather_object_list(obj) -> list:
    """Gather arbitrary picklable objects from all ranks onto every rank.

    pickle -> byte tensor -> pad-to-max -> all_gather (reference
    RetinaNet train_utils/distributed_utils.py:73-113)."""
    world = get_world_size()
    if world == 1:
        return [obj]
    device = torch.device("cuda", get_local_rank()) if (
        torch.cuda.is_available() and dist.get_backend() == "nccl") else torch.device("cpu")
    data = pickle.dumps(obj)
    t = torch.frombuffer(bytearray(data), dtype=torch.uint8).to(device)
    n = torch.tensor([t.numel()], dtype=torch.long, device=device)
    sizes = [torch.zeros(1, dtype=torch.long, device=device) for _ in range(world)]
    dist.all_gather(sizes, n)
    sizes = [int(s.item()) for s in sizes]
    max_size = max(sizes)
    if t.numel() < max_size:
        t = torch.cat([t, torch.zeros(max_size - t.numel(), dtype=torch.uint8, device=device)])
    out = [torch.zeros(max_size, dtype=torch.uint8, device=device) for _ in range(world)]
    dist.all_gather(out, t)
    res = []
    for s, o in zip(sizes, out):
        res.append(pickle.loads(o[:s].cpu().numpy().tobytes()))
    return res


@contextmanager
def zero_first():
    """Make all non-zero ranks wait for rank 0 (dataset prep / cache build),
    reference swin utils/torch_utils.py:16-23."""
    if is_dist() and get_rank() != 0:
        dist.barrier()
    yield
    if is_dist() and get_rank() == 0:
        dist.barrier()


def shared_random_seed() -> int:
    """One random seed agreed across all ranks (reference YOLOX utils/dist.py:268)."""
    import numpy as np

    seed = int(np.random.randint(2**31))
    if not is_dist():
        return seed
    t = torch.tensor([seed], dtype=torch.long)
    if torch.cuda.is_available() and dist.get_backend() == "nccl":
        t = t.cuda()
    dist.broadcast(t, src=0)
    return int(t.item())


def find_free_port() -> int:
    """Bind port 0 to let the OS pick a free port (reference YOLOX
    yolox/core/launch.py:24-36)."""
    import socket

    sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    sock.bind(("", 0))
    port = sock.getsockname()[1]
    sock.close()
    return port


def _launch_worker(local_rank, main_func, world_size, num_gpus, machine_rank,
                   dist_url, args):
    host_port = dist_url.split("//")[-1]
    host, port = host_port.rsplit(":", 1)
    os.environ.update(MASTER_ADDR=host, MASTER_PORT=port,
                      RANK=str(machine_rank * num_gpus + local_rank),
                      WORLD_SIZE=str(world_size), LOCAL_RANK=str(local_rank))
    main_func(*args)


def launch(main_func, num_gpus_per_machine, num_machines=1, machine_rank=0,
           dist_url="auto", args=(), start_method="spawn"):
    """Self-spawning multi-process launcher (reference YOLOX
    yolox/core/launch.py:39-147): free-port autodetect + mp.start_processes.

    Each worker exports the torchrun-style env (MASTER_*/RANK/LOCAL_RANK/
    WORLD_SIZE) and calls main_func, whose init_distributed() then joins the
    group over RCCL (or gloo on CPU). world_size==1 calls main_func inline.
    """
    import torch.multiprocessing as mp

    world_size = num_machines * num_gpus_per_machine
    if world_size <= 1:
        return main_func(*args)
    if dist_url == "auto":
        assert num_machines == 1, "dist_url=auto needs a single machine"
        dist_url = f"tcp://127.0.0.1:{find_free_port()}"
    mp.start_processes(
        _launch_worker, nprocs=num_gpus_per_machine,
        args=(main_func, world_size, num_gpus_per_machine, machine_rank,
              dist_url, args),
        daemon=False, start_method=start_method)
