"""Sharded checkpoint save/restore.

Capability parity: /root/reference/epl/runtime/saver.py — ShardingLoader
(restore with name remap + begin/size slicing for resharding, :46-128)
and MemoryEfficientBuilder (sharded, serialized shard writes, :145-207);
plus the hook behavior that only the first constructor saves replicated
variables while split variables are saved per shard
(/root/reference/epl/parallel/hooks.py:542-590, shard-suffixed names from
ops/distributed_dense.py:111-123).

Layout on disk (directory):
  meta.pt                         — world/topology/step info (rank 0)
  tg{T}_pos{P}.pt                 — taskgraph T's params owned by
                                    within-replica position P (replica 0
                                    only; replicated taskgraphs save one
                                    file from position 0).  Split shards
                                    keep their shard index in the file AND
                                    '{param}.shard{P}' keys.
  opt_rank{R}.pt                  — per-rank optimizer state (ZeRO keeps
                                    states local, reference hooks.py:340-344)
"""

import os

import torch


def _tg_params(tg):
    out = {}
    for name, mod in zip(tg.module_names, tg.modules):
        prefix = name + "." if name else ""
        for pn, p in mod.state_dict(keep_vars=True).items():
            out[prefix + pn] = p
    return out


def _save_tg_bucketed(tg, pos, nshards, path, bucket_bytes):
    """Stream one taskgraph's params to disk in ~bucket-sized part files:
    never more than one bucket of host copies alive at a time (reference
    MemoryEfficientBuilder's 50 MB bucketed serialized writes,
    /root/reference/epl/runtime/saver.py:145-207)."""
    params = _tg_params(tg)
    parts = 0
    cur, cur_bytes = {}, 0

    def flush():
        nonlocal parts, cur, cur_bytes
        if not cur:
            return
        torch.save(cur, os.path.join(
            path, "tg{}_pos{}.part{}.pt".format(tg.index, pos, parts)))
        parts += 1
        cur, cur_bytes = {}, 0

    shard_dims = {}
    for k, t in params.items():
        key = "{}.shard{}".format(k, pos) if tg.is_split else k
        cur[key] = t.detach().to("cpu")
        if tg.is_split and getattr(t, "_epl_shard_dim", None) is not None:
            shard_dims[k] = int(t._epl_shard_dim)
        cur_bytes += t.numel() * t.element_size()
        if cur_bytes >= bucket_bytes:
            flush()
    flush()
    torch.save(
        {"taskgraph": tg.index, "position": pos, "is_split": tg.is_split,
         "nshards": nshards, "nparts": parts, "shard_dims": shard_dims},
        os.path.join(path, "tg{}_pos{}.pt".format(tg.index, pos)))


def save_checkpoint(engine, path, save_optimizer=True):
    os.makedirs(path, exist_ok=True)
    rank = engine.rank
    if rank == 0:
        torch.save({
            "world_size": engine.world_size,
            "per_replica": engine.per_replica,
            "num_stages": engine.num_stages,
            "global_step": engine.global_step,
            "amp_loss_scale": engine.amp.loss_scale,
            "taskgraphs": [
                {"index": tg.index, "type": tg.strategy_type,
                 "device_count": tg.device_count,
                 "module_names": tg.module_names}
                for tg in engine.plan.taskgraphs],
        }, os.path.join(path, "meta.pt"))
    bucket_bytes = int(engine.config.io.checkpoint_bucket_mb) * 1024 * 1024
    serial = engine.config.io.serial_checkpoint_writes
    turns = range(engine.world_size) if serial else [None]
    for turn in turns:
        if turn is None or turn == rank:
            if engine.replica_id == 0:
                for tg in engine._owned_tgs:
                    ranks = tg.virtual_device.local_ranks(0)
                    pos = ranks.index(rank)
                    if not tg.is_split and pos != 0:
                        continue  # replicated: first position saves
                    _save_tg_bucketed(tg, pos,
                                      len(ranks) if tg.is_split else 1,
                                      path, bucket_bytes)
        if turn is not None:
            _barrier(engine)
    if save_optimizer:
        sd = engine.optimizer.state_dict()
        sd_cpu = _to_cpu(sd)
        torch.save(sd_cpu, os.path.join(path, "opt_rank{}.pt".format(rank)))
        if getattr(engine, "sparse_handlers", None):
            torch.save(
                [_to_cpu(h.state_dict()) for h in engine.sparse_handlers],
                os.path.join(path, "opt_sparse_rank{}.pt".format(rank)))
    _barrier(engine)


def _to_cpu(obj):
    if torch.is_tensor(obj):
        return obj.detach().to("cpu")
    if isinstance(obj, dict):
        return {k: _to_cpu(v) for k, v in obj.items()}
    if isinstance(obj, list):
        return [_to_cpu(v) for v in obj]
    return obj


def _barrier(engine):
    import torch.distributed as dist
    if dist.is_initialized():
        dist.barrier()


class ShardingLoader:
    """Restore with optional name remap and re-sharding: a split
    taskgraph saved with N shards can restore onto M shards — tensors are
    concatenated on their shard dim and re-sliced (begin/size semantics
    of the reference's sharding_info, runtime/saver.py:46-128)."""

    def __init__(self, path, assign_map=None):
        self.path = path
        self.assign_map = assign_map or {}
        self.meta = torch.load(os.path.join(path, "meta.pt"),
                               weights_only=False)

    def _remap(self, name):
        for old, new in self.assign_map.items():
            if name.startswith(old):
                return new + name[len(old):]
        return name

    def load_into(self, engine, load_optimizer=True, strict=True):
        for tg in engine._owned_tgs:
            ranks = tg.virtual_device.local_ranks(engine.replica_id)
            pos = ranks.index(engine.rank)
            if tg.is_split:
                self._load_split_tg(tg, pos, len(ranks), strict)
            else:
                self._load_replicated_tg(tg, strict)
        # params were loaded into module tensors == arena views; refresh
        # fp32 masters from the arenas FIRST — if optimizer state follows,
        # it restores the exact fp32 master (higher precision than the
        # bf16 params) and re-syncs params from it, so refreshing after
        # the optimizer load would round the master through bf16.
        for fg in engine.flat_groups:
            fg.refresh_master()
        for h in getattr(engine, "sparse_handlers", []):
            h.refresh_master()
        if load_optimizer:
            opt_path = os.path.join(
                self.path, "opt_rank{}.pt".format(engine.rank))
            if os.path.exists(opt_path):
                engine.optimizer.load_state_dict(
                    torch.load(opt_path, weights_only=False))
            sp_path = os.path.join(
                self.path, "opt_sparse_rank{}.pt".format(engine.rank))
            if os.path.exists(sp_path) and getattr(
                    engine, "sparse_handlers", None):
                blobs = torch.load(sp_path, weights_only=False)
                for h, b in zip(engine.sparse_handlers, blobs):
                    h.load_state_dict(b)
        _barrier(engine)

    def _head_and_blobs(self, tgindex, pos):
        """Yield the param dicts of tg/pos — one per bucketed part file,
        or the single legacy blob.  Returns (head, iterator)."""
        f = os.path.join(self.path, "tg{}_pos{}.pt".format(tgindex, pos))
        head = torch.load(f, weights_only=False)

        def gen():
            if "params" in head:   # legacy single-blob format
                yield head["params"]
                return
            for i in range(head.get("nparts", 0)):
                yield torch.load(os.path.join(
                    self.path, "tg{}_pos{}.part{}.pt".format(
                        tgindex, pos, i)), weights_only=False)
        return head, gen()

    def _merged_params(self, tgindex, pos):
        head, blobs = self._head_and_blobs(tgindex, pos)
        out = {}
        for b in blobs:
            out.update(b)
        head = dict(head)
        head["params"] = out
        return head

    def _load_replicated_tg(self, tg, strict):
        params = _tg_params(tg)
        want = {self._remap(k): (k, t) for k, t in params.items()}
        found = set()
        _, blobs = self._head_and_blobs(tg.index, 0)
        for blob in blobs:   # one bucket of host tensors at a time
            for src, t in blob.items():
                if src in want:
                    want[src][1].data.copy_(
                        t.to(want[src][1].device, want[src][1].dtype))
                    found.add(src)
        if strict:
            missing = set(want) - found
            if missing:
                raise KeyError(
                    "missing checkpoint tensor(s) {}".format(sorted(missing)))

    def _load_split_tg(self, tg, pos, nshards, strict):
        # gather available shard files (reshard path merges each
        # position's parts; the same-count fast path below could stream,
        # but resharding needs all shards of a tensor at once anyway)
        import glob
        files = sorted(glob.glob(os.path.join(
            self.path, "tg{}_pos*.pt".format(tg.index))))
        files = [f for f in files if ".part" not in os.path.basename(f)]
        shards = [
            self._merged_params(
                tg.index,
                torch.load(f, weights_only=False)["position"])
            for f in files]
        saved_n = shards[0]["nshards"] if shards else 0
        params = _tg_params(tg)
        for k, t in params.items():
            base = self._remap(k)
            if saved_n == nshards:
                key = "{}.shard{}".format(base, pos)
                blob = shards[pos]["params"]
                if key in blob and tuple(blob[key].shape) == tuple(t.shape):
                    t.data.copy_(blob[key].to(t.device, t.dtype))
                    continue
                if key not in blob and strict:
                    raise KeyError(key)
                # shape mismatch: same shard count but a different shard
                # LAYOUT (e.g. column-sharded checkpoint restoring into a
                # row-parallel module under auto_pair_sequential) — fall
                # through to the full reshard path below
            # reshard: concatenate all shards on the SAVED shard dim,
            # re-slice on the target's shard dim with the
            # remainder-to-shard-0 policy
            pieces = []
            saved_dim = None
            for s in shards:
                key = "{}.shard{}".format(base, s["position"])
                if key in s["params"]:
                    pieces.append(s["params"][key])
                    if saved_dim is None:
                        saved_dim = s.get("shard_dims", {}).get(base)
            if not pieces:
                if strict:
                    raise KeyError(base)
                continue
            tdim = getattr(t, "_epl_shard_dim", None)
            tdim = 0 if tdim is None else tdim
            if saved_dim is None:
                # the saved tensor was REPLICATED across positions (e.g.
                # a row-parallel bias): any piece is the full tensor
                full = pieces[0]
            else:
                full = torch.cat(pieces, dim=saved_dim)
            if tuple(full.shape) == tuple(t.shape):
                t.data.copy_(full.to(t.device, t.dtype))
                continue
            from easyparallellibrary_amd.ops.distributed_dense import (
                shard_offset, shard_size)
            total = full.shape[tdim]
            lo = shard_offset(total, nshards, pos)
            n = shard_size(total, nshards, pos)
            t.data.copy_(
                full.narrow(tdim, lo, n).to(t.device, t.dtype))


def load_checkpoint(engine, path, load_optimizer=True, assign_map=None,
                    strict=True):
    ShardingLoader(path, assign_map).load_into(engine, load_optimizer,
                                               strict)
