"""One-sided (async) table over hipIpc-mapped peer shards.

Reference: the ET remote-access path (RemoteAccessOpSender/Handler +
CommManager) lets any worker pull/push any key at any time —
no bulk synchrony. The collective data plane (et/comm.py) is the
throughput path; THIS is its async analogue, the MI355X way:

  * each rank's shard lives in hipMalloc memory (NOT the torch caching
    allocator — IPC needs the allocation base) and is exported once with
    `hipIpcGetMemHandle` (dmabuf mode: HSA_ENABLE_IPC_MODE_LEGACY=0);
  * every peer maps every shard once at setup; a PULL is then just a
    gather kernel dereferencing the peer pointer (peer HBM over xGMI,
    p2p loads), and a PUSH is an atomicAdd scatter kernel — no message,
    no rendezvous, no matching collective on the owner;
  * consistency is per-element atomic adds (the owner-side `add` update
    function algebra); richer update functions need the collective path.

This makes SSP slack > 0 a REAL bounded-async mode: workers proceed at
their own pace (uneven batch counts, no deadlock) with the SSP clock as
the only cross-worker coupling. v1 scope: float32 dense tables, `add`
update fn. v2 adds owner-side apply-queue rings for arbitrary update
fns. v3 adds LIVE MIGRATION (`migrate_blocks`): adopters read moving
blocks straight out of the owner's HBM (one-sided, no owner
participation in the data move), then every rank remaps at a
generation-versioned quiesce point — the ownership-first protocol of
MigrationExecutor.java:48 expressed as drain -> read -> remap.
"""

from __future__ import annotations

import torch

from harmony_amd.config import TableConfig
from harmony_amd.et.table import Table


def _require_hip():
    from harmony_amd import ops

    hip = ops._load_hip()
    if hip is None or not torch.cuda.is_available():
        raise RuntimeError("one-sided tables need the HIP extension + a GPU")
    return hip


class OneSidedTable(Table):
    """Async-access dense table. Construct on every rank, then `connect()`
    collectively (exchanges IPC handles through the control store)."""

    # update fns applied by direct system-scope atomics (v1 path):
    #   add        — atomic float/int add IS the update
    #   lda_counts — ±1 count deltas; the reference's clamp>=0 is a
    #                defensive no-op (counts conserved by construction,
    #                docs/ROADMAP.md), so atomic int add is exact
    _ATOMIC_FNS = ("add", "lda_counts")

    def __init__(self, cfg: TableConfig, rank: int, world_size: int,
                 device: torch.device, store=None,
                 ring_capacity: int = 8192):
        from harmony_amd import ops as _ops

        # NON-add update fns (NMF's clamp(old - step*delta), assign) go
        # through owner-side apply-queue RINGS (v2, ops/csrc/os_ring.hip):
        # writers enqueue (key, delta) asynchronously; the owner alone
        # drains and applies — the reference's per-block op-queue write
        # serialization (CommManager.java:36-155).
        self._ring_mode = cfg.update_fn not in self._ATOMIC_FNS
        if self._ring_mode:
            assert _ops.fused_apply_supported(cfg.update_fn), \
                f"one-sided v2 needs a device apply mode for {cfg.update_fn}"
            assert cfg.dtype == "float32", "ring payloads are f32"
        else:
            assert cfg.dtype in ("float32", "int32")
        self._hip = _require_hip()
        super().__init__(cfg, rank, world_size, device)
        self.store = store
        self._ring_cap = int(ring_capacity)
        self._peer_ptr = {}          # rank -> mapped device pointer (int)
        self._ring_peer = {}         # rank -> mapped RING pointer (int)
        if self._ring_mode:
            vd = cfg.value_dim
            nwords = (self._hip.os_ring_bytes(world_size, self._ring_cap,
                                              vd) + 3) // 4
            self._ring_buf = self._hip.os_shard_alloc(nwords, 1, 0)
            # host-side exact backpressure state (we are the only writer
            # of our slot in each peer's ring)
            self._pushed = {r: 0 for r in range(world_size)}
            self._head_cache = {r: 0 for r in range(world_size)}
            self._scratch = torch.zeros(2, dtype=torch.int64, device=device)
        # per-table op stats (reference RemoteAccessOpStat): pulled rows /
        # pushed rows / remote bytes moved over xGMI
        self.stats = {"pull_rows": 0, "push_rows": 0, "remote_bytes": 0}
        self._gen = 0                # IPC-mapping generation (bumped by
        #                              migrate_blocks' collective remap)
        import threading

        # serializes THIS process's async ops against migrate_blocks'
        # unmap/remap window (e.g. CachedOneSidedAccessor's background
        # refresh thread pulling while peers are closed). RLock: ring
        # push's backpressure wait re-enters drain().
        self._lk = threading.RLock()
        # jobserver fail-fast flag (runtime/control.py ControlPlane): ring
        # backpressure and migration barriers poll it so a dead peer
        # unwinds this rank instead of wedging it for the full timeout
        self._failed_key = "js/failed"
        # every rank's block->slot map is derivable from the static
        # round-robin ownership, so remote row indices need no exchange
        self._peer_slot = {}
        for r in range(world_size):
            slot = torch.full((cfg.num_blocks,), -1, dtype=torch.int64)
            for i, b in enumerate(self.ownership.owned_blocks(r)):
                slot[b] = i
            self._peer_slot[r] = slot.to(device)

    def _check_failed(self) -> None:
        from harmony_amd.runtime.control import JobCancelled

        try:
            failed = self.store is not None \
                and self.store.check([self._failed_key])
        except Exception:            # noqa: BLE001 — store gone == failed
            failed = True
        if failed:
            raise JobCancelled("one-sided wait: jobserver failed fast")

    # ------------------------------------------------------------- lifecycle

    def _alloc_shard(self) -> None:
        # hipMalloc-backed shard (IPC-exportable base), then the normal
        # deterministic per-block init copied in
        owned = self.ownership.owned_blocks(self.rank)
        self._local_blocks = list(owned)
        rows = len(owned) * self.part.block_size
        self.shard = self._hip.os_shard_alloc(
            rows, self.cfg.value_dim, 1 if self.cfg.dtype == "int32" else 0)
        from harmony_amd.et import update_functions as uf
        from harmony_amd.et.table import _block_seed

        init = uf.init_fn(self.cfg.init_fn)
        bs = self.part.block_size
        for i, b in enumerate(owned):
            t = init((bs, self.cfg.value_dim), self.dtype, self.device,
                     seed=_block_seed(self.cfg.table_id, b),
                     **self.cfg.init_args)
            self.shard[i * bs:(i + 1) * bs] = t
        slot = torch.full((self.cfg.num_blocks,), -1, dtype=torch.int64)
        for i, b in enumerate(owned):
            slot[b] = i
        self._block_slot = slot.to(self.device)

    def connect(self, store=None) -> None:
        """Collective: export my shard (+ ring), map every peer's. Keys are
        generation-versioned so a re-connect after migration can never read
        a peer's pre-migration (stale) handle."""
        store = store or self.store
        assert store is not None
        torch.cuda.synchronize()
        key = f"os/{self.cfg.table_id}/g{self._gen}"
        h = self._hip.os_ipc_handle(self.shard)
        store.set(f"{key}/{self.rank}", bytes(h.tolist()).hex())
        if self._ring_mode:
            hr = self._hip.os_ipc_handle(self._ring_buf)
            store.set(f"{key}/ring/{self.rank}", bytes(hr.tolist()).hex())
        for r in range(self.world_size):
            if r == self.rank:
                continue
            raw = store.get(f"{key}/{r}")
            hb = torch.tensor(list(bytes.fromhex(raw.decode())),
                              dtype=torch.uint8)
            self._peer_ptr[r] = self._hip.os_ipc_open(hb)
            if self._ring_mode:
                raw = store.get(f"{key}/ring/{r}")
                hb = torch.tensor(list(bytes.fromhex(raw.decode())),
                                  dtype=torch.uint8)
                self._ring_peer[r] = self._hip.os_ipc_open(hb)

    def close(self) -> None:
        with self._lk:
            self._close_locked()

    def _close_locked(self) -> None:
        for p in self._peer_ptr.values():
            self._hip.os_ipc_close(p)
        self._peer_ptr.clear()
        for p in self._ring_peer.values():
            self._hip.os_ipc_close(p)
        self._ring_peer.clear()

    # ------------------------------------------------------------ async ops

    def _owner_of(self, blocks: torch.Tensor) -> torch.Tensor:
        if getattr(self, "_owner_dev", None) is None:
            self._owner_dev = self.ownership.owner.to(self.device,
                                                      torch.int64)
        return self._owner_dev[blocks]

    def _rows_on(self, r: int, keys: torch.Tensor) -> torch.Tensor:
        blocks = self.part.block_of(keys)
        return (self._peer_slot[r][blocks] * self.part.block_size
                + self.part.offset_in_block(keys))

    def pull(self, keys: torch.Tensor) -> torch.Tensor:
        """Async pull: gather rows straight out of each owner's HBM."""
        with self._lk:
            return self._pull_locked(keys)

    def _pull_locked(self, keys: torch.Tensor) -> torch.Tensor:
        keys = keys.to(self.device, torch.int64)
        self.stats["pull_rows"] += keys.shape[0]
        owner = self._owner_of(self.part.block_of(keys))
        out = torch.empty((keys.shape[0], self.cfg.value_dim),
                          dtype=self.dtype, device=self.device)
        for r in range(self.world_size):
            sel = owner == r
            if not bool(sel.any()):
                continue
            rows = self._rows_on(r, keys[sel])
            if r != self.rank:
                self.stats["remote_bytes"] += (rows.shape[0]
                                               * self.cfg.value_dim * 4)
            # local reads go through the same system-scope gather kernel:
            # a plain torch read can hit a line this process cached before
            # a peer's system-scope atomic landed
            ptr = (self.shard.data_ptr() if r == self.rank
                   else self._peer_ptr[r])
            out[sel] = self._hip.os_gather(
                ptr, rows, self.cfg.value_dim,
                1 if self.cfg.dtype == "int32" else 0)
        return out

    def push(self, keys: torch.Tensor, deltas: torch.Tensor) -> None:
        """Async push. add-algebra fns: atomicAdd scatter into each
        owner's HBM (v1). Other fns: enqueue into each owner's apply-queue
        ring (v2) — the owner applies on its next drain()."""
        with self._lk:
            self._push_locked(keys, deltas)

    def _push_locked(self, keys: torch.Tensor,
                     deltas: torch.Tensor) -> None:
        keys = keys.to(self.device, torch.int64)
        self.stats["push_rows"] += keys.shape[0]
        deltas = deltas.to(self.device, self.dtype).contiguous()
        owner = self._owner_of(self.part.block_of(keys))
        for r in range(self.world_size):
            sel = owner == r
            if not bool(sel.any()):
                continue
            k = keys[sel]
            d = deltas[sel].contiguous()
            if r != self.rank:
                self.stats["remote_bytes"] += (k.shape[0]
                                               * self.cfg.value_dim * 4)
            if self._ring_mode:
                if r == self.rank:
                    # owner-local fast path: the owner is the ONLY applier,
                    # and this thread is the owner's applier thread, so a
                    # direct apply preserves the per-block serialization
                    self._apply_local(k, d)
                else:
                    self._ring_push_to(r, k, d)
                continue
            rows = self._rows_on(r, k)
            # local pushes use the SAME system-scope atomic kernel as
            # remote ones: torch index_add_ is a plain read-modify-write,
            # and racing it against another process's atomics on the same
            # cells can drop updates
            ptr = (self.shard.data_ptr() if r == self.rank
                   else self._peer_ptr[r])
            self._hip.os_scatter_add(ptr, rows, d)

    # --------------------------------------------------- v2 ring plumbing

    def _apply_local(self, keys: torch.Tensor, deltas: torch.Tensor) -> None:
        """Apply update_fn(shard[key], delta) IN ORDER. Duplicate keys in
        one batch are applied in occurrence rounds (scatter_apply assumes
        unique rows; a concurrent read-modify-write on duplicates would
        lose updates) — f(f(v,d1),d2), the reference's sequential op-queue
        semantics."""
        from harmony_amd import ops as _ops

        ua = self.cfg.update_args or {}
        kw = dict(step_size=float(ua.get("step_size", 0.0)),
                  max_val=float(ua.get("max_val", 0.0)))
        rows = self._rows_on(self.rank, keys)
        order = torch.argsort(rows, stable=True)
        sr = rows[order]
        # occurrence index of each item within its key group
        uniq, inv, counts = torch.unique_consecutive(
            sr, return_inverse=True, return_counts=True)
        starts = torch.zeros_like(counts)
        starts[1:] = counts.cumsum(0)[:-1]
        occ = torch.arange(sr.numel(), device=sr.device) - starts[inv]
        max_occ = int(counts.max()) if counts.numel() else 0
        if max_occ <= 1:
            _ops.scatter_apply(self.shard, rows, deltas,
                               self.cfg.update_fn, **kw)
            return
        sd = deltas[order]
        for o in range(max_occ):
            sel = occ == o
            _ops.scatter_apply(self.shard, sr[sel], sd[sel],
                               self.cfg.update_fn, **kw)

    def _ring_push_to(self, r: int, keys: torch.Tensor,
                      deltas: torch.Tensor) -> None:
        import time

        n = int(keys.shape[0])
        cap, vd = self._ring_cap, self.cfg.value_dim
        if n > cap:
            # a push larger than the ring: send in capacity-sized chunks
            # (each chunk applies backpressure; per-writer FIFO holds)
            for off in range(0, n, cap):
                self._ring_push_to(r, keys[off:off + cap],
                                   deltas[off:off + cap])
            return
        # exact host-side backpressure: we are the only writer of our slot
        deadline = time.monotonic() + 30.0
        it = 0
        while self._pushed[r] + n - self._head_cache[r] > cap:
            it += 1
            if it % 100 == 0:
                self._check_failed()
            # refresh the owner's head (remote read), bounded wait
            self._hip.os_ring_read_head(self._ring_peer[r], self.world_size,
                                        cap, vd, self.rank, self._scratch)
            torch.cuda.synchronize()
            self._head_cache[r] = int(self._scratch[0])
            if self._pushed[r] + n - self._head_cache[r] <= cap:
                break
            # drain OUR ring while waiting: two ranks pushing large
            # batches at each other would otherwise deadlock in mutual
            # backpressure before either reaches its own drain
            self.drain()
            if time.monotonic() > deadline:
                raise RuntimeError(
                    f"one-sided ring to rank {r} full for 30s "
                    "(owner not draining?)")
            time.sleep(0.001)
        self._hip.os_ring_reserve(self._ring_peer[r], self.world_size, cap,
                                  vd, self.rank, n, self._scratch[1:])
        self._hip.os_ring_push(self._ring_peer[r], self.world_size, cap, vd,
                               self.rank, self._scratch[1:],
                               keys.contiguous(), deltas)
        self._pushed[r] += n

    def drain(self, max_per: int = 0) -> int:
        """OWNER-side: apply queued remote pushes in per-writer order with
        this table's update function (call between batches; the reference's
        per-block op queue drains continuously on comm threads — here the
        batch boundary is the natural quiesce point). Returns items applied."""
        if not self._ring_mode:
            return 0
        with self._lk:
            return self._drain_locked(max_per)

    def _drain_locked(self, max_per: int = 0) -> int:
        cap, vd = self._ring_cap, self.cfg.value_dim
        max_per = max_per or cap
        keys, deltas, counts = self._hip.os_ring_drain(
            self._ring_buf.data_ptr(), self.world_size, cap, vd, max_per)
        counts = counts.cpu()
        total = 0
        for w in range(self.world_size):
            c = int(counts[w])
            if not c:
                continue
            o = w * max_per
            self._apply_local(keys[o:o + c], deltas[o:o + c])
            total += c
        return total

    def pull_full(self) -> torch.Tensor:
        """Async full-table pull (dense apps: MLR/Lasso pull everything)."""
        keys = torch.arange(self.cfg.num_keys, device=self.device)
        return self.pull(keys)

    def fence(self) -> None:
        """Make my issued pushes visible device-wide before a clock tick."""
        torch.cuda.synchronize()

    # -------------------------------------------- inherited-API guard rails

    def update(self, keys, deltas, assume_unique: bool = False) -> None:
        # add-algebra fns -> atomic push; other fns -> ring enqueue (v2)
        self.push(keys, deltas)

    def push_dense(self, grad_full) -> None:
        self.push(torch.arange(self.cfg.num_keys, device=self.device),
                  grad_full)

    def get(self, keys):
        return self.pull(keys)

    def pull_all(self):
        return self.pull_full()

    def drop_blocks(self, blocks) -> None:
        raise RuntimeError("one-sided shards move only through the "
                           "collective migrate_blocks (IPC mappings pin "
                           "the layout between remap generations)")

    def adopt_blocks(self, blocks) -> None:
        raise RuntimeError("one-sided shards move only through the "
                           "collective migrate_blocks (IPC mappings pin "
                           "the layout between remap generations)")

    # ------------------------------------------------------- live migration

    def _os_barrier(self, tag: str) -> None:
        """Store-counter barrier (no collective plane on async jobs)."""
        import time

        key = f"os/{self.cfg.table_id}/bar/g{self._gen}/{tag}"
        self.store.add(key, 1)
        deadline = time.monotonic() + 120.0
        it = 0
        while int(self.store.add(key, 0)) < self.world_size:
            it += 1
            if it % 200 == 0:
                self._check_failed()
            if time.monotonic() > deadline:
                raise RuntimeError(f"one-sided barrier {tag} timed out")
            time.sleep(0.0005)

    def migrate_blocks(self, moves) -> None:
        """Collective live migration (every rank, identical `moves` =
        {block_id: dst_rank}, at a quiesced point — PlanExecutor calls it
        between batches). Ownership-first protocol, one-sided data move:

        1. quiesce — everyone fences issued pushes, owners drain rings;
        2. adopters gather the moving blocks' rows DIRECTLY from the
           current owner's HBM over xGMI (os_gather on the mapped peer
           pointer; the owner does nothing);
        3. remap — unmap all peers, flip ownership, rebuild the local
           shard in a fresh IPC-exportable allocation (kept blocks copied
           device-to-device, adopted blocks from step 2), reset rings,
           re-export under the next generation and re-map.

        Reference: MigrationExecutor.java:48 ownership-first migration;
        the per-block access locks collapse to this phase-level quiesce."""
        with self._lk:
            self._migrate_locked(moves)

    def _migrate_locked(self, moves) -> None:
        moves = {int(b): int(d) for b, d in dict(moves).items()
                 if self.ownership.owner_of_int(int(b)) != int(d)}
        if self.world_size == 1 or not moves:
            self.ownership.update_many(moves)
            return
        assert self.store is not None, "migrate_blocks needs the store"
        bs, vd = self.part.block_size, self.cfg.value_dim
        isint = 1 if self.cfg.dtype == "int32" else 0
        # 1. quiesce: all pushes issued + visible, all rings applied
        self.fence()
        self._os_barrier("q0")
        self.drain()
        torch.cuda.synchronize()
        self._os_barrier("q1")
        # 2. one-sided reads of adopted blocks from the (pre-flip) owner
        adopted = {}
        for b, dst in moves.items():
            if dst != self.rank:
                continue
            src = self.ownership.owner_of_int(b)
            rows = (int(self._peer_slot[src][b]) * bs
                    + torch.arange(bs, device=self.device))
            ptr = (self.shard.data_ptr() if src == self.rank
                   else self._peer_ptr[src])
            adopted[b] = self._hip.os_gather(ptr, rows, vd, isint)
            self.stats["remote_bytes"] += bs * vd * 4
        torch.cuda.synchronize()
        self._os_barrier("read")         # owners keep shards until here
        # 3. remap: close peer mappings, flip ownership, rebuild shard
        self.close()
        self._os_barrier("closed")
        self.ownership.update_many(moves)
        self._owner_dev = None
        old_shard = self.shard
        old_slot = {b: i for i, b in enumerate(self._local_blocks)}
        new_owned = list(self.ownership.owned_blocks(self.rank))
        new_shard = self._hip.os_shard_alloc(len(new_owned) * bs, vd, isint)
        for i, b in enumerate(new_owned):
            dstv = new_shard[i * bs:(i + 1) * bs]
            if b in old_slot:
                j = old_slot[b]
                dstv.copy_(old_shard[j * bs:(j + 1) * bs])
            else:
                dstv.copy_(adopted[b])
        self.shard = new_shard
        self._local_blocks = new_owned
        slot = torch.full((self.cfg.num_blocks,), -1, dtype=torch.int64)
        for i, b in enumerate(new_owned):
            slot[b] = i
        self._block_slot = slot.to(self.device)
        for r in range(self.world_size):
            s = torch.full((self.cfg.num_blocks,), -1, dtype=torch.int64)
            for i, b in enumerate(self.ownership.owned_blocks(r)):
                s[b] = i
            self._peer_slot[r] = s.to(self.device)
        if self._ring_mode:
            # rings are empty (drained at the quiesce); restart counters
            self._ring_buf.zero_()
            self._pushed = {r: 0 for r in range(self.world_size)}
            self._head_cache = {r: 0 for r in range(self.world_size)}
        torch.cuda.synchronize()
        del old_shard, adopted           # freed only after "closed" barrier
        self._gen += 1
        self.connect()                   # re-export + re-map at gen+1
