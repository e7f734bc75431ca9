"""Patient-shard data parallelism over RCCL/xGMI (or gloo on CPU).

The MI355X-native replacement for the reference's process-level sharding
(one producer thread per patient, Kafka partitions keyed by patient id —
SURVEY.md §2.5): patient streams are sharded across the 8 GPUs of a node by
consistent key hash, each rank runs its own StreamEngine + MyCNNEngine over
its shard, and per-trigger prediction batches are all-gathered over xGMI so
every rank (and the store writer) sees the full prediction set.

xGMI topology note: prediction payloads are tiny (4 B per
stream), so the all-gather is latency- not bandwidth-bound; a single direct
all-gather (one hop on each of the 7 p2p links) is the right collective —
torch.distributed.all_gather over RCCL emits exactly that for same-size
tensors.
"""

from __future__ import annotations

import os
from typing import List, Optional, Sequence, Tuple

import torch
import torch.distributed as dist


def pick_backend(world: int) -> str:
    """RCCL (backend "nccl" on ROCm) when every rank can own a GPU —
    the production one-process-per-GPU topology. When ranks outnumber
    visible devices (world>1 validation on a 1-GPU box), RCCL refuses with
    "Duplicate GPU detected", so fall back to gloo: compute stays on the
    GPU, collectives stage through host memory (tiny payloads — see
    all_gather_predictions)."""
    if not torch.cuda.is_available():
        return "gloo"
    return "nccl" if torch.cuda.device_count() >= world else "gloo"


def init_distributed(backend: Optional[str] = None) -> Tuple[int, int]:
    """Initialize from torchrun env; returns (rank, world). No-op world=1."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 0, 1
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        if backend is None:
            backend = pick_backend(world)
        dist.init_process_group(backend)
    return dist.get_rank(), dist.get_world_size()


def _fnv1a(s: str) -> int:
    h = 1469598103934665603
    for ch in s.encode():
        h = ((h ^ ch) * 1099511628211) & 0xFFFFFFFFFFFFFFFF
    return h


def shard_for_key(key: str, world: int) -> int:
    """Consistent patient->rank assignment (stable across restarts; same
    FNV-1a the bus uses for partition keying)."""
    return _fnv1a(key) % world if world > 1 else 0


def shard_streams(keys: Sequence[str], rank: int, world: int) -> List[str]:
    return [k for k in keys if shard_for_key(k, world) == rank]


def all_gather_predictions(probs: torch.Tensor,
                           group=None) -> torch.Tensor:
    """All-gather same-size per-rank prediction vectors -> (world, S).

    Over RCCL this is the direct small-message all-gather on xGMI; on gloo
    (CPU tests) the same call sites work unchanged.
    """
    if not dist.is_initialized() or dist.get_world_size(group) == 1:
        return probs.unsqueeze(0)
    world = dist.get_world_size(group)
    backend = dist.get_backend(group)
    if backend == "gloo" and probs.is_cuda:
        # gloo has no CUDA all_gather: stage through host (validation-only
        # topology — ranks sharing one GPU; production uses RCCL direct)
        host = probs.contiguous().cpu()
        out = [torch.empty_like(host) for _ in range(world)]
        dist.all_gather(out, host, group=group)
        return torch.stack(out).to(probs.device)
    out = [torch.empty_like(probs) for _ in range(world)]
    dist.all_gather(out, probs.contiguous(), group=group)
    return torch.stack(out)


def all_reduce_max(value: float, device=None) -> float:
    if not dist.is_initialized():
        return value
    if dist.get_backend() == "gloo":
        device = "cpu"  # gloo reduces host tensors
    t = torch.tensor([value], dtype=torch.float64,
                     device=device or "cpu")
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())


class DPServing:
    """One rank's serving shard: StreamEngine + MyCNNEngine + all-gather.

    BASELINE.json config 3: 8 patient-stream shards DP across 8 x MI355X,
    RCCL all-gather of predictions over xGMI.
    """

    def __init__(self, streams_per_rank: int, model=None, device: str = "cpu",
                 n_channels: int = 10, ring_grid: int = 2048,
                 fs: float = 125.0, dtype: torch.dtype = torch.bfloat16):
        from tskd_amd.engine import StreamEngine
        from tskd_amd.models import build_model
        from tskd_amd.ops import MyCNNEngine
        self.rank, self.world = (dist.get_rank(), dist.get_world_size()) \
            if dist.is_initialized() else (0, 1)
        self.device = torch.device(device)
        self.dtype = dtype if self.device.type == "cuda" else torch.float32
        self.se = StreamEngine(streams_per_rank, n_channels,
                               ring_grid=ring_grid, fs=fs, device=device)
        self.me = MyCNNEngine(model or build_model("MyCNN5").eval(),
                              device=device)
        self.age = torch.full((streams_per_rank, 1), 65.0, device=self.device)

    def step(self, raw: torch.Tensor,
             chan_map: Optional[Sequence[int]] = None) -> torch.Tensor:
        """Ingest one trigger of raw data for this shard, score every stream,
        all-gather predictions. Returns (world, S) probabilities."""
        self.se.ingest_dense(raw, chan_map=chan_map)
        w = self.se.windows(batch=1, stride=12, dtype=self.dtype)
        probs = self.me.forward(w, self.age, apply_sigmoid=True)
        return all_gather_predictions(probs.reshape(-1))
