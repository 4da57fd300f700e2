"""Data-parallel gradient communication over RCCL/xGMI.

The reference has no distributed code at all; this module is the
MI355X-native addition. Design decisions for one 8-GPU xGMI node (7 p2p links
× ≈153 GB/s per GPU):

- one process per GPU, `torch.distributed` with backend "nccl" (= RCCL on
  ROCm); CPU tests use "gloo" with the same code path.
- dense grads (TRANSFORM, ATTENTION, and the full-softmax target-table grad)
  are all-reduced asynchronously the moment they are produced, so the big
  target-table reduction (≈200 MB bf16 on java14m) overlaps the rest of
  backward (see Code2VecNetwork.train_step ordering).
- embedding grads are NEVER dense-all-reduced (the tables are 666+466 MB):
  each rank DEDUPES AND SUMS its (row-id, grad-row) pairs, then ranks
  all-gather the unique rows (~31 MB/rank at Zipf-real ids vs 157 MB raw;
  async, overlapped with the dense Adam chain). Shard counts and the
  ragged-tail termination votes travel on a host-side gloo group (no
  device work, no stream fences). Padding rows replicate a real id with
  zero-valued grads — a no-op under the hash-dedup lazy Adam.
- C2V_DP_SPARSE=owner switches to the owner-sharded reduce: id%N
  all-to-all (per-link parallel), owner-side reduce, disjoint-shard gather.
- the big target-table all-reduce runs on its own communicator so it never
  head-of-line-blocks the sparse gathers.
- scalar metrics are averaged with a tiny all-reduce.
"""

import os
from typing import Dict, Optional, Tuple

import torch
import torch.distributed as dist


class _PendingSparseGather:
    """Handle for in-flight sparse-grad all-gathers: callers enqueue
    independent device work (the dense w/a Adam chain) between launch and
    wait(), so the comm overlaps compute instead of fencing the stream."""

    def __init__(self, pending):
        self._pending = pending

    def wait(self):
        results = []
        for h1, h2, ids_out, rows_out in self._pending:
            h1.wait()
            h2.wait()
            results.append((torch.cat(ids_out), torch.cat(rows_out)))
        return results


def init_distributed_from_env(backend: Optional[str] = None) -> Tuple[int, int]:
    """Initialize torch.distributed from torchrun env vars if present.
    Returns (rank, world_size); (0, 1) when not distributed."""
    world_size = int(os.environ.get('WORLD_SIZE', '1'))
    if world_size <= 1:
        return 0, 1
    if not dist.is_initialized():
        if backend is None:
            backend = 'nccl' if torch.cuda.is_available() else 'gloo'
        rank = int(os.environ.get('RANK', '0'))
        local_rank = int(os.environ.get('LOCAL_RANK', rank))
        if backend == 'nccl':
            torch.cuda.set_device(local_rank)
        dist.init_process_group(backend=backend, rank=rank, world_size=world_size)
    return dist.get_rank(), dist.get_world_size()


class Reducer:
    """Gradient reducer for one training step. Keys identify parameters; dense
    reductions are launched async and waited for right before the matching
    Adam step."""

    def __init__(self, process_group=None, assume_equal_shards: bool = False):
        """assume_equal_shards: skip the per-step shard-size exchange (and its
        host syncs) when every rank contributes identically-shaped sparse
        grads — true whenever the per-rank batch size is fixed (the bench and
        steady-state training; ragged tails need False). Only meaningful for
        the raw (non-dedup) gather path: deduped counts always differ."""
        self.group = process_group
        self.assume_equal_shards = assume_equal_shards
        self._handles: Dict[str, object] = {}
        self._host_group = None
        self._bulk_group = None
        self._count_pin = None

    def _host_pg(self):
        """Process group for tiny host-side exchanges (sparse shard counts,
        termination votes). Under RCCL these run on a separate gloo group so
        they neither enqueue device work nor force a stream sync; under gloo
        the main group already is host-side."""
        if dist.get_backend(self.group) != 'nccl':
            return self.group
        if self._host_group is None:
            self._host_group = dist.new_group(backend='gloo')
        return self._host_group

    def _bulk_pg(self):
        """Dedicated group (own RCCL communicator + stream) for the big
        target-table all-reduce: torch serializes collectives of one group
        on one stream, and at DP=8 the 2.3 ms ring all-reduce would
        otherwise head-of-line-block the sparse gathers whose consumer
        runs much earlier than the all-reduce's."""
        if self._bulk_group is None:
            self._bulk_group = dist.new_group()
        return self._bulk_group

    @property
    def world_size(self) -> int:
        if not dist.is_initialized():
            return 1
        return dist.get_world_size(self.group)

    def allreduce_dense(self, key: str, tensor: torch.Tensor):
        if self.world_size <= 1:
            return tensor
        group = self._bulk_pg() if key == 'target_table' else self.group
        handle = dist.all_reduce(tensor, op=dist.ReduceOp.SUM, group=group,
                                 async_op=True)
        self._handles[key] = (handle, tensor)
        return tensor

    def wait(self, key: str):
        if key in self._handles:
            handle, tensor = self._handles.pop(key)
            handle.wait()
            tensor.div_(self.world_size)

    def allgather_sparse(self, ids: torch.Tensor,
                         rows: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """All-gather (ids, rows) sparse gradient contributions. All ranks in
        lock-step batches have identical row counts here (fixed B×C per rank),
        but the code pads defensively for ragged tails. Gathered rows are
        averaged by world size to match dense all-reduce semantics."""
        ws = self.world_size
        if ws <= 1:
            return ids, rows
        if self.assume_equal_shards:
            n = ids.numel()
            ids_out = [torch.empty_like(ids) for _ in range(ws)]
            rows_out = [torch.empty_like(rows) for _ in range(ws)]
            dist.all_gather(ids_out, ids.contiguous(), group=self.group)
            dist.all_gather(rows_out, rows.contiguous(), group=self.group)
            return torch.cat(ids_out), torch.cat(rows_out) / ws
        n_local = torch.tensor([ids.numel()], dtype=torch.int64, device=ids.device)
        counts = [torch.zeros_like(n_local) for _ in range(ws)]
        dist.all_gather(counts, n_local, group=self.group)
        counts = [int(c.item()) for c in counts]
        n_max = max(counts)
        if ids.numel() < n_max:
            pad = n_max - ids.numel()
            # replicate a real id with zero grads: no-op under lazy sparse Adam
            ids = torch.cat([ids, ids.new_full((pad,), int(ids[0].item()))])
            rows = torch.cat([rows, rows.new_zeros((pad, rows.shape[1]))])
        ids_out = [torch.empty_like(ids) for _ in range(ws)]
        rows_out = [torch.empty_like(rows) for _ in range(ws)]
        dist.all_gather(ids_out, ids.contiguous(), group=self.group)
        dist.all_gather(rows_out, rows.contiguous(), group=self.group)
        all_ids = torch.cat([t[:c] for t, c in zip(ids_out, counts)])
        all_rows = torch.cat([t[:c] for t, c in zip(rows_out, counts)])
        return all_ids, all_rows / ws

    def allgather_sparse_dedup(self, entries):
        """All-gather rank-locally deduped (unique ids, summed rows) sparse
        gradients. `entries` is a list of (uniq_ids, acc_rows, count) triples
        as returned by ops.functional.sparse_dedup_sum*; all entries share
        ONE shard-count exchange. Returns a list of (ids, rows) with the DP
        mean (1/ws) already folded into the gathered rows.

        Wire format: rows travel bf16 on GPU (halves the xGMI bytes; the
        fp32 local sums are rounded once) and fp32 on CPU (exactness tests);
        ranks pad to the max count with a REAL id (their first unique id)
        carrying all-zero rows — a no-op contribution under the hash-dedup
        sparse Adam, so no slicing is needed after the gather."""
        ws = self.world_size
        assert ws > 1
        dev = entries[0][0].device
        cuda = dev.type == 'cuda'
        # local counts to host: one batched async D2H + event (GPU), no-op (CPU)
        if cuda:
            cnt_dev = torch.cat([c.reshape(-1).to(torch.int64)
                                 for _, _, c in entries])
            if self._count_pin is None or self._count_pin.numel() < cnt_dev.numel():
                self._count_pin = torch.empty(cnt_dev.numel(),
                                              dtype=torch.int64,
                                              pin_memory=True)
            pin = self._count_pin[:cnt_dev.numel()]
            pin.copy_(cnt_dev, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record()
            ev.synchronize()      # waits for the dedup kernels only: later
            local = pin.tolist()  # device work keeps flowing behind this
        else:
            local = [int(c) if not torch.is_tensor(c) else int(c.item())
                     for _, _, c in entries]
        # count exchange on the host group (no device work, ~µs)
        t = torch.tensor(local, dtype=torch.int64)
        counts = [torch.empty_like(t) for _ in range(ws)]
        dist.all_gather(counts, t, group=self._host_pg())
        pending = []
        for j, (ids, rows, _) in enumerate(entries):
            c = local[j]
            n_max = max(int(cr[j]) for cr in counts)
            wire_dt = torch.bfloat16 if cuda else rows.dtype
            wire_ids = torch.empty(n_max, dtype=torch.int64, device=dev)
            wire_rows = torch.empty(n_max, rows.shape[1], dtype=wire_dt,
                                    device=dev)
            if c > 0:
                wire_ids[:c] = ids[:c]
                wire_rows[:c] = rows[:c] * (1.0 / ws)
            pad_id = ids[0] if c > 0 else torch.zeros(
                (), dtype=torch.int64, device=dev)
            if n_max > c:
                wire_ids[c:] = pad_id
                wire_rows[c:] = 0
            ids_out = [torch.empty_like(wire_ids) for _ in range(ws)]
            rows_out = [torch.empty_like(wire_rows) for _ in range(ws)]
            h1 = dist.all_gather(ids_out, wire_ids, group=self.group,
                                 async_op=True)
            h2 = dist.all_gather(rows_out, wire_rows, group=self.group,
                                 async_op=True)
            pending.append((h1, h2, ids_out, rows_out))
        return _PendingSparseGather(pending)

    def _all_to_all_rows(self, send_ids, send_rows):
        """Exchange per-destination (ids, rows) lists: returns concatenated
        receives. NCCL runs true all-to-all over xGMI (per-link parallel);
        gloo (CPU tests) emulates with an all-gather + own-slice select."""
        ws = self.world_size
        rank = dist.get_rank(self.group)
        if dist.get_backend(self.group) == 'nccl':
            send_id_sizes = [t.numel() for t in send_ids]
            # exchange sizes on the host group
            sz = torch.tensor(send_id_sizes, dtype=torch.int64)
            all_sz = [torch.empty_like(sz) for _ in range(ws)]
            dist.all_gather(all_sz, sz, group=self._host_pg())
            recv_sizes = [int(all_sz[src][rank]) for src in range(ws)]
            d = send_rows[0].shape[1]
            dev = send_rows[0].device
            recv_ids = [torch.empty(n, dtype=torch.int64, device=dev)
                        for n in recv_sizes]
            recv_rows = [torch.empty(n, d, dtype=send_rows[0].dtype,
                                     device=dev) for n in recv_sizes]
            dist.all_to_all(recv_ids, list(send_ids), group=self.group)
            dist.all_to_all(recv_rows, list(send_rows), group=self.group)
            return torch.cat(recv_ids), torch.cat(recv_rows)
        # gloo: all-gather every rank's full send set, keep what's ours
        mine_ids, mine_rows = [], []
        for dst in range(ws):
            ids_d, rows_d = send_ids[dst], send_rows[dst]
            n_local = torch.tensor([ids_d.numel()], dtype=torch.int64)
            counts = [torch.zeros_like(n_local) for _ in range(ws)]
            dist.all_gather(counts, n_local, group=self.group)
            n_max = max(int(c.item()) for c in counts)
            pad_ids = torch.zeros(n_max, dtype=torch.int64)
            pad_ids[:ids_d.numel()] = ids_d
            pad_rows = torch.zeros(n_max, send_rows[0].shape[1],
                                   dtype=rows_d.dtype)
            pad_rows[:rows_d.shape[0]] = rows_d
            out_i = [torch.empty_like(pad_ids) for _ in range(ws)]
            out_r = [torch.empty_like(pad_rows) for _ in range(ws)]
            dist.all_gather(out_i, pad_ids, group=self.group)
            dist.all_gather(out_r, pad_rows, group=self.group)
            if dst == rank:
                for src in range(ws):
                    c = int(counts[src].item())
                    mine_ids.append(out_i[src][:c])
                    mine_rows.append(out_r[src][:c])
        return torch.cat(mine_ids), torch.cat(mine_rows)

    def reduce_sparse_owner(self, entries):
        """Owner-sharded sparse reduction (opt-in alternative to the dedup
        all-gather, C2V_DP_SPARSE=owner): each rank sends its deduped rows
        to owner rank `id % N` (all-to-all: per-link parallel over xGMI),
        owners sum their shard, and the DISJOINT reduced sets are
        all-gathered. Wire bytes ~ V_local*(N-1)/N spread over all links
        plus V_global/N per rank on the gather — at java14m Zipf shapes
        roughly half the dedup-gather's ring bytes (profiles/
        r02_dp_volume.md). Returns [(ids, rows)] like
        allgather_sparse_dedup(...).wait()."""
        from ..ops import functional as F
        ws = self.world_size
        assert ws > 1
        results = []
        for ids, rows, cnt in entries:
            if torch.is_tensor(cnt):
                c = int(cnt.item())
            else:
                c = int(cnt)
            ids_l = ids[:c]
            rows_l = rows[:c] * (1.0 / ws)
            if ids_l.is_cuda:
                rows_l = rows_l.to(torch.bfloat16)
            owner = (ids_l % ws).to(torch.int64)
            send_ids, send_rows = [], []
            for dst in range(ws):
                sel = (owner == dst).nonzero(as_tuple=True)[0]
                send_ids.append(ids_l.index_select(0, sel))
                send_rows.append(rows_l.index_select(0, sel))
            got_ids, got_rows = self._all_to_all_rows(send_ids, send_rows)
            # owner-side reduce of the shard (ids collide across ranks)
            if got_ids.numel():
                ru, racc, rc = F.sparse_dedup_sum(got_ids, got_rows)
                rc = int(rc.item()) if torch.is_tensor(rc) else int(rc)
                red_ids, red_rows = ru[:rc], racc[:rc]
                if ids_l.is_cuda:
                    red_rows = red_rows.to(torch.bfloat16)
            else:
                red_ids = got_ids
                red_rows = got_rows
            # gather the disjoint reduced shards
            n_local = torch.tensor([red_ids.numel()], dtype=torch.int64)
            counts = [torch.zeros_like(n_local) for _ in range(ws)]
            dist.all_gather(counts, n_local, group=self._host_pg())
            n_max = max(int(x.item()) for x in counts)
            dev = red_ids.device
            # padding uses id -1 (never a vocab id); consumers filter it
            # out after the gather — unlike the dedup-gather path there is
            # no guaranteed local id to replicate (a shard can be empty)
            wire_ids = torch.full((n_max,), -1, dtype=torch.int64,
                                  device=dev)
            wire_rows = torch.zeros(n_max, rows_l.shape[1],
                                    dtype=red_rows.dtype, device=dev)
            if red_ids.numel():
                wire_ids[:red_ids.numel()] = red_ids
                wire_rows[:red_rows.shape[0]] = red_rows
            ids_out = [torch.empty_like(wire_ids) for _ in range(ws)]
            rows_out = [torch.empty_like(wire_rows) for _ in range(ws)]
            dist.all_gather(ids_out, wire_ids, group=self.group)
            dist.all_gather(rows_out, wire_rows, group=self.group)
            all_ids = torch.cat(ids_out)
            all_rows = torch.cat(rows_out)
            keep = (all_ids >= 0).nonzero(as_tuple=True)[0]
            if keep.numel() != all_ids.numel():
                all_ids = all_ids.index_select(0, keep)
                all_rows = all_rows.index_select(0, keep)
            results.append((all_ids, all_rows))
        return results

    # -- termination consensus --------------------------------------------
    # Ranks can end an epoch with unequal batch counts (the DP shard split is
    # not batch-aligned); a rank stepping once more than its peers would
    # deadlock in the gradient collectives. Each step every rank votes
    # whether it has a NEXT batch; training continues only while ALL do
    # (surplus batches on longer ranks are dropped). The vote is posted
    # asynchronously on the host group one step ahead, so it rides along the
    # step's compute instead of adding a blocking latency tax per step.

    def start_vote(self, have_next: bool):
        if self.world_size <= 1:
            return have_next
        t = torch.tensor([1 if have_next else 0], dtype=torch.int32)
        handle = dist.all_reduce(t, op=dist.ReduceOp.MIN,
                                 group=self._host_pg(), async_op=True)
        return handle, t

    def finish_vote(self, vote) -> bool:
        if self.world_size <= 1:
            return bool(vote)
        handle, t = vote
        handle.wait()
        return bool(int(t.item()))

    def all_continue(self, have_next: bool) -> bool:
        """Blocking vote (start+finish back to back) — kept for callers
        without a lookahead batch."""
        return self.finish_vote(self.start_vote(have_next))

    def allreduce_mean_scalar(self, value: float) -> float:
        if self.world_size <= 1:
            return value
        t = torch.tensor([value], dtype=torch.float64)
        if dist.get_backend(self.group) == 'nccl':
            t = t.cuda()
        dist.all_reduce(t, op=dist.ReduceOp.SUM, group=self.group)
        return float(t.item()) / self.world_size
