"""Distributed primitives: RCCL (torch.distributed backend "nccl" == RCCL on
ROCm) over xGMI, with a gloo fallback for CPU tests.

MI355X topology note: each GPU pair on one node has its own
xGMI link (7 links x ~153 GB/s), so direct pairwise all-to-all saturates all
links at once — the join shuffle below exchanges each rank pair directly.
"""
from __future__ import annotations

import os
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist


def init_from_env(device_hint: Optional[str] = None) -> Tuple[int, int, torch.device]:
    """Initialize torch.distributed from torchrun env vars; returns
    (rank, world, device).  Single-process (no env) => (0, 1, device)."""
    if "RANK" not in os.environ or int(os.environ.get("WORLD_SIZE", "1")) <= 1:
        dev = torch.device(device_hint or ("cuda:0" if torch.cuda.is_available() else "cpu"))
        return 0, 1, dev
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
        dev = torch.device(f"cuda:{local_rank}")
        backend = "nccl"
    else:
        dev = torch.device("cpu")
        backend = "gloo"
    if not dist.is_initialized():
        dist.init_process_group(backend=backend)
    return rank, world, dev


def is_dist() -> bool:
    return dist.is_initialized() and dist.get_world_size() > 1


def barrier():
    if is_dist():
        dist.barrier()


_AR_BUF = {}


def allreduce_sum_scalar(x: int, device) -> int:
    if not is_dist():
        return x
    # cached per-device buffer: the serving hot loop calls this per step
    t = _AR_BUF.get(device)
    if t is None:
        t = _AR_BUF[device] = torch.zeros(1, dtype=torch.int64,
                                          device=device)
    t.fill_(x)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return int(t.item())


def _alltoall_sizes(send_counts: torch.Tensor) -> torch.Tensor:
    """Exchange per-peer row counts; returns recv_counts."""
    world = dist.get_world_size()
    recv = torch.zeros_like(send_counts)
    dist.all_to_all_single(recv, send_counts) if dist.get_backend() == "nccl" \
        else _gloo_alltoall_sizes(recv, send_counts)
    return recv


def _gloo_alltoall_sizes(recv: torch.Tensor, send: torch.Tensor):
    world = dist.get_world_size()
    gathered = [torch.zeros_like(send) for _ in range(world)]
    dist.all_gather(gathered, send)
    rank = dist.get_rank()
    for peer in range(world):
        recv[peer] = gathered[peer][rank]


def all_to_all_rows(cols: List[torch.Tensor], dest: torch.Tensor
                    ) -> List[torch.Tensor]:
    """Repartition a row table: row i goes to rank dest[i].

    The K-shuffle of the distributed join: buckets are
    staged contiguously per destination and exchanged pairwise — on RCCL
    this is grouped send/recv over per-pair xGMI links; on gloo (CPU tests)
    it falls back to isend/irecv pairs.
    """
    if not is_dist():
        return list(cols)
    world = dist.get_world_size()
    device = cols[0].device
    # bucket rows by destination (stable sort keeps determinism)
    order = torch.argsort(dest, stable=True)
    sorted_cols = [c[order] for c in cols]
    sorted_dest = dest[order]
    send_counts = torch.bincount(sorted_dest, minlength=world).to(torch.int64)
    recv_counts = _alltoall_sizes(send_counts.cpu())
    out_cols: List[torch.Tensor] = []
    backend = dist.get_backend()
    send_list = send_counts.cpu().tolist()
    recv_list = recv_counts.cpu().tolist()
    n_out = int(sum(recv_list))
    for c in sorted_cols:
        out = torch.empty(n_out, dtype=c.dtype, device=device)
        if backend == "nccl":
            dist.all_to_all_single(out, c.contiguous(),
                                   output_split_sizes=recv_list,
                                   input_split_sizes=send_list)
        else:
            _p2p_exchange(out, c.contiguous(), send_list, recv_list)
        out_cols.append(out)
    return out_cols


def _p2p_exchange(out: torch.Tensor, inp: torch.Tensor,
                  send_list: List[int], recv_list: List[int]):
    """gloo fallback: pairwise isend/irecv with self-copy."""
    rank = dist.get_rank()
    world = dist.get_world_size()
    send_offs = [0]
    for c in send_list:
        send_offs.append(send_offs[-1] + c)
    recv_offs = [0]
    for c in recv_list:
        recv_offs.append(recv_offs[-1] + c)
    reqs = []
    for peer in range(world):
        if peer == rank:
            out[recv_offs[peer]:recv_offs[peer + 1]] = \
                inp[send_offs[peer]:send_offs[peer + 1]]
            continue
        if recv_list[peer] > 0:
            reqs.append(dist.irecv(out[recv_offs[peer]:recv_offs[peer + 1]], src=peer))
        if send_list[peer] > 0:
            reqs.append(dist.isend(inp[send_offs[peer]:send_offs[peer + 1]].contiguous(), dst=peer))
    for r in reqs:
        r.wait()


def all_gather_rows(cols: List[torch.Tensor]) -> List[torch.Tensor]:
    """Replicate a small row table on every rank (broadcast join build
    side): variable-size all_gather per column."""
    if not is_dist():
        return list(cols)
    world = dist.get_world_size()
    device = cols[0].device
    n = torch.tensor([cols[0].numel()], dtype=torch.int64,
                     device=device if dist.get_backend() == "nccl" else "cpu")
    sizes = [torch.zeros_like(n) for _ in range(world)]
    dist.all_gather(sizes, n)
    sizes = [int(s.item()) for s in sizes]
    max_n = max(sizes) if sizes else 0
    out: List[torch.Tensor] = []
    for c in cols:
        pad = torch.zeros(max_n, dtype=c.dtype, device=device)
        pad[:c.numel()] = c
        gathered = [torch.zeros_like(pad) for _ in range(world)]
        dist.all_gather(gathered, pad)
        out.append(torch.cat([g[:sizes[i]] for i, g in enumerate(gathered)]))
    return out
