"""Collective data plane: pull/push as RCCL collectives over xGMI.

Replaces the reference's Avro RPC remote-access path (RemoteAccessOpSender/
Handler, CommManager — reference et/evaluator/impl/*.java, message schema
elastictable.avsc:74-128) with bucketed collectives, the idiomatic shape for
one-process-per-GPU over xGMI's point-to-point mesh:

  multiGetOrInit(all keys)   -> all-gather of per-rank shards       (dense apps)
  multiGetOrInit(sparse keys)-> all-to-all-v of key ids + gathered rows
  multiUpdate(dense delta)   -> reduce-scatter + fused update epilogue
  multiUpdate(sparse deltas) -> local segment-sum, all-to-all-v of
                                (keys, deltas), owner-side scatter-apply

Every op is collective within the job's process group: all ranks of a job
enter the same op in the same order (enforced by the task-unit scheduler,
runtime/control.py). On CPU/gloo (tests) all-to-all is emulated with batched
isend/irecv, which gloo supports.
"""

from __future__ import annotations

import itertools
from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

# identity for route-cacheable key tensors: a fresh id per marked tensor —
# data_ptr() could alias a freed tensor's reallocated address (advisor r01)
_route_ids = itertools.count(1)


class DataPlane:
    def __init__(self, group: Optional[dist.ProcessGroup], rank: int,
                 world_size: int, device: torch.device):
        self.group = group
        self.rank = rank
        self.world_size = world_size
        self.device = device
        self.backend = dist.get_backend(group) if dist.is_initialized() else "none"
        self._perm_cache: Dict[Tuple[str, int], torch.Tensor] = {}
        # routing cache for STATIC key sets (a mini-batch block's keys never
        # change, so the per-pull argsort/bincount is paid once per
        # (block, ownership version) instead of every batch)
        self._route_cache: Dict[tuple, tuple] = {}
        # recv-count cache for static routes: ranks advance through batch
        # blocks in lockstep (collective phases), so the full W x W count
        # matrix of a (block, ownership-version) pair is static too — caching
        # it removes one all-gather + host sync per phase
        self._counts_cache: Dict[tuple, tuple] = {}

    def invalidate_routes(self) -> None:
        """Drop route + recv-count caches. MUST be called collectively
        whenever any rank's key sets change out-of-band (batch reslicing
        by an elasticity plan): a per-rank cache hit/miss divergence would
        put ranks in different stages of the same push (observed hang),
        and cached recv counts would be stale even when consistent."""
        self._route_cache.clear()
        self._counts_cache.clear()

    # ------------------------------------------------------------------ utils

    def _supports_a2a(self) -> bool:
        return self.backend == "nccl"

    def _all_to_all_v(self, send: torch.Tensor, send_splits: List[int],
                      recv_splits: List[int]) -> torch.Tensor:
        """All-to-all with per-rank element counts on dim 0."""
        recv_shape = (sum(recv_splits),) + tuple(send.shape[1:])
        recv = torch.empty(recv_shape, dtype=send.dtype, device=send.device)
        if self._supports_a2a():
            dist.all_to_all_single(recv, send.contiguous(),
                                   output_split_sizes=recv_splits,
                                   input_split_sizes=send_splits,
                                   group=self.group)
            return recv
        # gloo emulation: batched isend/irecv (deterministic order).
        send_parts = list(torch.split(send, send_splits, dim=0))
        recv_parts = list(torch.split(recv, recv_splits, dim=0))
        reqs = []
        for peer in range(self.world_size):
            if peer == self.rank:
                recv_parts[peer].copy_(send_parts[peer])
                continue
            if recv_splits[peer] > 0:
                reqs.append(dist.irecv(recv_parts[peer], src=peer, group=self.group))
            if send_splits[peer] > 0:
                reqs.append(dist.isend(send_parts[peer].contiguous(), dst=peer,
                                       group=self.group))
        for r in reqs:
            r.wait()
        return recv

    def _exchange_counts(self, counts: torch.Tensor
                         ) -> Tuple[torch.Tensor, int]:
        """counts[i] = #elems this rank sends to rank i. All-gathers the
        FULL W x W count matrix (tiny) and returns (recv counts for me,
        global total) — the total lets callers skip a globally-empty data
        exchange SYMMETRICALLY on every rank (a per-rank skip would strand
        the other ranks inside the collective). NCCL/RCCL requires device
        tensors; split sizes must be host ints, so results return on CPU."""
        dev = self.device if self.backend == "nccl" else torch.device("cpu")
        counts = counts.to(dev)
        gathered = [torch.empty_like(counts) for _ in range(self.world_size)]
        dist.all_gather(gathered, counts, group=self.group)
        mat = torch.stack(gathered)
        return (mat[:, self.rank].contiguous().cpu(), int(mat.sum()))

    def _gather_perm(self, table) -> torch.Tensor:
        """Permutation p with full_rows[p] = concat(shards in rank order):
        maps gathered rows to their global key positions."""
        key = (table.cfg.table_id, table.ownership.version)
        perm = self._perm_cache.get(key)
        if perm is None:
            bs = table.block_size
            idx = []
            for r in range(self.world_size):
                for b in table.ownership.owned_blocks(r):
                    idx.append(torch.arange(b * bs, (b + 1) * bs, dtype=torch.int64))
            perm = torch.cat(idx).to(self.device)
            self._perm_cache = {key: perm}  # keep only current version
        return perm

    def _contiguous_even(self, table) -> bool:
        """True iff ownership is the initial contiguous even partition AND
        divides evenly (enables zero-copy all-gather / reduce-scatter)."""
        B, W = table.cfg.num_blocks, self.world_size
        if B % W:
            return False
        per = B // W
        own = table.ownership.owner
        expected = torch.arange(W, dtype=torch.int32).repeat_interleave(per)
        return bool(torch.equal(own, expected))

    # --------------------------------------------------------------- pull all

    def pull_all(self, table) -> torch.Tensor:
        """Gather the full table -> [padded_num_keys, value_dim]."""
        vdim = table.cfg.value_dim
        n_full = table.cfg.padded_num_keys
        if self._contiguous_even(table) and self.backend == "nccl":
            out = torch.empty((n_full, vdim), dtype=table.dtype, device=self.device)
            dist.all_gather_into_tensor(out, table.shard, group=self.group)
            return out
        # General ownership: gather variable-size shards (padded) + permute.
        counts = [c * table.block_size for c in table.ownership.counts()]
        mx = max(counts) if counts else 0
        send = table.shard
        if send.shape[0] < mx:
            send = torch.cat([send, torch.zeros((mx - send.shape[0], vdim),
                                                dtype=table.dtype, device=self.device)])
        bufs = [torch.empty((mx, vdim), dtype=table.dtype, device=self.device)
                for _ in range(self.world_size)]
        dist.all_gather(bufs, send.contiguous(), group=self.group)
        rows = torch.cat([bufs[r][:counts[r]] for r in range(self.world_size)])
        out = torch.empty((n_full, vdim), dtype=table.dtype, device=self.device)
        out[self._gather_perm(table)] = rows
        return out

    # -------------------------------------------------------------- push dense

    def push_dense(self, table, grad_full: torch.Tensor) -> None:
        """Reduce-scatter a full-table delta; fused update on owned rows."""
        grad_full = grad_full.to(table.dtype)
        if self._contiguous_even(table) and self.backend == "nccl":
            out = torch.empty_like(table.shard)
            dist.reduce_scatter_tensor(out, grad_full.contiguous(), group=self.group)
            table.apply_update_dense_local(out)
            return
        dist.all_reduce(grad_full, group=self.group)
        perm = self._gather_perm(table)
        # local segment of the permutation = rows of our shard
        counts = [c * table.block_size for c in table.ownership.counts()]
        start = sum(counts[:self.rank])
        mine = perm[start:start + counts[self.rank]]
        table.apply_update_dense_local(grad_full[mine])

    # -------------------------------------------------------------- pull keys

    def _route(self, table, keys: torch.Tensor):
        """Sort keys by owner rank; returns (sorted_keys, order, send_splits).

        Routing is cached only for tensors explicitly marked long-lived
        (`keys._harmony_static = True`, set by batch constructors): caching by
        data_ptr alone would alias freed/reallocated tensors."""
        cacheable = getattr(keys, "_harmony_static", False)
        ck = None
        if cacheable:
            rid = getattr(keys, "_harmony_route_id", None)
            if rid is None:
                rid = next(_route_ids)
                keys._harmony_route_id = rid
            ck = (table.cfg.table_id, rid, table.ownership.version)
        if ck is not None:
            hit = self._route_cache.get(ck)
            if hit is not None:
                return hit
        owners = table.ownership.owner.to(keys.device)[table.part.block_of(keys)]
        order = torch.argsort(owners.to(torch.int64), stable=True)
        sorted_keys = keys[order]
        splits = torch.bincount(owners.to(torch.int64),
                                minlength=self.world_size)
        out = (sorted_keys, order, splits, ck)
        if ck is not None:
            if len(self._route_cache) > 512:
                self._route_cache.clear()
            self._route_cache[ck] = out
        return out

    def _counts_for(self, send_counts: torch.Tensor, ck):
        """Count exchange with caching for static routes (see __init__)."""
        if ck is not None:
            hit = self._counts_cache.get(ck)
            if hit is not None:
                return hit
        out = self._exchange_counts(send_counts)
        if ck is not None:
            if len(self._counts_cache) > 512:
                self._counts_cache.clear()
            self._counts_cache[ck] = out
        return out

    def pull_keys(self, table, keys: torch.Tensor) -> torch.Tensor:
        keys = keys.to(self.device, torch.int64)
        sorted_keys, order, send_counts, ck = self._route(table, keys)
        recv_counts, total = self._counts_for(send_counts, ck)
        ssp, rsp = send_counts.tolist(), recv_counts.tolist()
        if total == 0:
            return torch.empty((0, table.cfg.value_dim), dtype=table.dtype,
                               device=self.device)
        req_keys = self._all_to_all_v(sorted_keys, ssp, rsp)      # keys we serve
        served = table.get_local(req_keys)                        # [n_req, vdim]
        vals_sorted = self._all_to_all_v(served, rsp, ssp)        # back to askers
        out = torch.empty_like(vals_sorted)
        out[order] = vals_sorted
        return out

    # -------------------------------------------------------------- push keys

    def push_keys(self, table, keys: torch.Tensor, deltas: torch.Tensor,
                  assume_unique: bool = False,
                  piggyback: Optional[torch.Tensor] = None
                  ) -> Optional[torch.Tensor]:
        """Push key/deltas to owners. `piggyback`: a small int vector
        folded into the count exchange and returned GLOBALLY SUMMED — lets
        callers fuse a tiny all-reduce (e.g. Pregel's halt vote) into the
        push's existing collective instead of issuing another one."""
        from harmony_amd.et.update_functions import merge_key_deltas

        keys = keys.to(self.device, torch.int64)
        deltas = deltas.to(self.device)
        # Aggregate locally first (reference CommManager serializes per-block
        # writes; merging before the wire preserves update semantics because
        # every registered update function is delta-merge associative — the
        # merge algebra (sum/min/last) comes from the update fn's MERGE_MODE).
        if assume_unique:
            uniq, agg = keys, deltas     # caller already aggregated per key
        else:
            uniq, agg = merge_key_deltas(keys, deltas, table.cfg.update_fn)
        sorted_keys, order, send_counts, ck = self._route(table, uniq)
        sorted_deltas = agg[order]
        pig_sums = None
        if piggyback is None:
            recv_counts, total = self._counts_for(send_counts, ck)
        else:
            # piggyback varies per call -> bypass the static-counts cache
            W = self.world_size
            dev = (self.device if self.backend == "nccl"
                   else torch.device("cpu"))
            # counts ride on the keys' device, the piggyback on the
            # caller's (often CPU): unify BEFORE cat — mixed-device cat
            # only crashed on the RCCL path (tests/test_comm_nccl1.py)
            ext = torch.cat([send_counts.to(dev, torch.int64),
                             piggyback.to(dev, torch.int64)])
            bufs = [torch.empty_like(ext) for _ in range(W)]
            dist.all_gather(bufs, ext, group=self.group)
            mat = torch.stack(bufs).cpu()
            recv_counts = mat[:, :W][:, self.rank].contiguous()
            total = int(mat[:, :W].sum())
            pig_sums = mat[:, W:].sum(0)
        if total == 0:
            return pig_sums
        ssp, rsp = send_counts.tolist(), recv_counts.tolist()
        recv_keys = self._all_to_all_v(sorted_keys, ssp, rsp)
        recv_deltas = self._all_to_all_v(sorted_deltas, ssp, rsp)
        if recv_keys.numel() == 0:
            return pig_sums
        # Aggregate across source ranks, then one update-fn apply per key.
        u2, agg2 = merge_key_deltas(recv_keys, recv_deltas,
                                    table.cfg.update_fn)
        table.update_local(u2, agg2)
        return pig_sums

    # ------------------------------------------------------------------ put

    def put_keys(self, table, keys: torch.Tensor, values: torch.Tensor) -> None:
        """multiPut (reference TableImpl.multiPut:156): route (key, value)
        pairs to owners; owners overwrite (no update function). Duplicate
        keys within one call: last writer wins (any is valid)."""
        keys = keys.to(self.device, torch.int64)
        values = values.to(self.device)
        sorted_keys, order, send_counts, ck = self._route(table, keys)
        sorted_vals = values[order]
        recv_counts, total = self._counts_for(send_counts, ck)
        if total == 0:
            return
        ssp, rsp = send_counts.tolist(), recv_counts.tolist()
        recv_keys = self._all_to_all_v(sorted_keys, ssp, rsp)
        recv_vals = self._all_to_all_v(sorted_vals, ssp, rsp)
        if recv_keys.numel():
            table.put_local(recv_keys, recv_vals)

    def remove_keys(self, table, keys: torch.Tensor) -> None:
        """multiRemove (reference TableImpl.remove:513): owners reset the
        rows to their deterministic init values (see Table.remove)."""
        keys = keys.to(self.device, torch.int64)
        sorted_keys, order, send_counts, ck = self._route(table, keys)
        recv_counts, total = self._counts_for(send_counts, ck)
        if total == 0:
            return
        ssp, rsp = send_counts.tolist(), recv_counts.tolist()
        recv_keys = self._all_to_all_v(sorted_keys, ssp, rsp)
        if recv_keys.numel():
            table.remove_local(recv_keys)

    # ------------------------------------------------------------ pair push

    def push_pairs(self, table, keys: torch.Tensor,
                   payload: torch.Tensor, apply_fn) -> None:
        """Route (key, payload-row) pairs to owners; the owner applies them
        with apply_fn(table, local_keys, local_payload). Used for compressed
        delta formats (LDA TopicChanges: payload = (old_topic, new_topic))
        where a dense per-key row would waste xGMI bandwidth."""
        keys = keys.to(self.device, torch.int64)
        sorted_keys, order, send_counts, ck = self._route(table, keys)
        sorted_payload = payload[order]
        recv_counts, total = self._counts_for(send_counts, ck)
        if total == 0:
            return
        ssp, rsp = send_counts.tolist(), recv_counts.tolist()
        recv_keys = self._all_to_all_v(sorted_keys, ssp, rsp)
        recv_payload = self._all_to_all_v(sorted_payload, ssp, rsp)
        if recv_keys.numel():
            apply_fn(table, recv_keys, recv_payload)

    # ---------------------------------------------------------- object tables

    def gather_equal(self, t: torch.Tensor) -> List[torch.Tensor]:
        """All-gather one tensor of IDENTICAL shape per rank; returns the
        per-rank list. The tensorized wire for object tables whose per-rank
        contribution count is fixed by construction (GBT: every rank
        pushes exactly one tree per class per batch) — replaces the
        pickled all_gather_object."""
        dev = self.device if self.backend == "nccl" else torch.device("cpu")
        t = t.to(dev).contiguous()
        bufs = [torch.empty_like(t) for _ in range(self.world_size)]
        dist.all_gather(bufs, t, group=self.group)
        return bufs

    def object_pull_all(self, table) -> Dict[int, object]:
        """Gather every key/value of an object table to every rank."""
        local = {}
        for blk in table.blocks.values():
            local.update(blk)
        bufs: List[Optional[dict]] = [None] * self.world_size
        dist.all_gather_object(bufs, local, group=self.group)
        merged: Dict[int, object] = {}
        for d in bufs:
            merged.update(d or {})
        return merged

    def object_push(self, table, items: List[Tuple[int, object]]) -> None:
        """Collective push: each rank contributes (key, delta) items; the
        owner of each key applies the table's update function."""
        bufs: List[Optional[list]] = [None] * self.world_size
        dist.all_gather_object(bufs, items, group=self.group)
        my_blocks = set(table.blocks.keys())
        for contrib in bufs:
            for key, delta in contrib or []:
                if table.part.block_of_int(key) in my_blocks:
                    table.update_local(key, delta)
