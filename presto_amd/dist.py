"""Multi-GPU exchange — the RCCL-over-xGMI analog of Presto's repartition
seam (PartitionedOutputOperator -> OutputBuffer -> HTTP pull ->
ExchangeOperator, SURVEY.md §3 stack 4): partition ids computed on-GPU with
the reference's partition math (fixed128.h pg_partition), bucketed columnar
split by the PARTITION operator, then torch.distributed all_to_all_single
(backend "nccl" == RCCL on ROCm) of the column buffers — no serialize/HTTP
hop.  Backend-agnostic (gloo on CPU for tests)."""
import torch
import torch.distributed as dist


def exchange_split_counts(send_counts):
    """all_to_all of row counts: returns recv_counts (rows arriving from
    each rank)."""
    world = dist.get_world_size()
    send = torch.tensor(send_counts, dtype=torch.int64)
    allc = [torch.zeros(world, dtype=torch.int64) for _ in range(world)]
    dist.all_gather(allc, send)
    me = dist.get_rank()
    return [int(allc[r][me]) for r in range(world)]


def exchange_columns(cols, send_counts, recv_counts=None, device=None):
    """cols: dict name -> 1-D tensor laid out partition-major (rows for rank
    0 first, then rank 1, ...), with send_counts[r] rows per destination.
    Returns dict of received tensors (concatenated in source-rank order,
    stable within each source rank)."""
    if recv_counts is None:
        recv_counts = exchange_split_counts(send_counts)
    out = {}
    n_recv = sum(recv_counts)
    for name, t in cols.items():
        r = torch.empty(n_recv, dtype=t.dtype,
                        device=device if device is not None else t.device)
        dist.all_to_all_single(r, t.contiguous(),
                               output_split_sizes=recv_counts,
                               input_split_sizes=list(send_counts))
        out[name] = r
    return out
