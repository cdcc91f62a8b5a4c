"""Pipeline-parallel runtime: micro-batch schedules over RCCL p2p.

Capability parity: /root/reference/epl/strategies/scheduler.py — the three
schedules PreferForward (GPipe, :36-50), PreferBackward (1F1B, :53-84, the
default) and PreferBackwardOptimizer (:87-116) — plus
graph_editor.py:610-668 (micro-batch gradient accumulation before DP
allreduce).

MI355X redesign: the reference encodes schedules as control-dependency
wiring between cloned per-(stage, micro-batch) subgraphs.  Here each
schedule is an imperative loop: activations/gradients move between
adjacent stages as RCCL send/recv over one xGMI hop (stage neighbours are
adjacent ranks by the replica-contiguous Layout), batched into one nccl
group per steady-state exchange so the bidirectional pair cannot deadlock.
Gradient accumulation falls out of p.grad views into the flat arena; the
DP allreduce hooks stay disabled until the schedule's final backward.
"""

import torch
import torch.distributed as dist

from easyparallellibrary_amd import constant
from easyparallellibrary_amd.comm.backend import create_communicator


class PipelineRuntime:
    def __init__(self, engine):
        self.engine = engine
        self.S = engine.num_stages
        self.s = engine.my_stage
        replica = engine.replica_id
        # Stages may span k>1 ranks each (e.g. EP-MoE or TP inside a
        # stage).  Position p of stage s exchanges activations with
        # position p of stage s±1 — k independent pipeline chains per
        # replica.  All stages must share one width.
        rank_lists = [tg.virtual_device.local_ranks(replica)
                      for tg in engine.stage_tgs]
        widths = {len(r) for r in rank_lists}
        self.mixed = len(widths) != 1
        if self.mixed:
            self._init_mixed(engine, rank_lists)
            return
        self.width = widths.pop()
        self.pos = rank_lists[self.s].index(engine.rank)
        self.stage_ranks = [r[self.pos] for r in rank_lists]
        self.prev_rank = self.stage_ranks[self.s - 1] if self.s > 0 else None
        self.next_rank = (self.stage_ranks[self.s + 1]
                          if self.s < self.S - 1 else None)
        # one p2p communicator per (replica, position) chain.  gloo group
        # creation is collective over ALL ranks, so every rank walks every
        # chain in the same order and keeps its own.
        self.comm = None
        for rep in range(engine.num_replicas):
            for p in range(self.width):
                chain = [tg.virtual_device.local_ranks(rep)[p]
                         for tg in engine.stage_tgs]
                mine = engine.rank in chain
                if not mine and (torch.cuda.is_available()
                                 or not dist.is_initialized()):
                    continue
                comm = create_communicator(
                    "{}_pipe_rep{}_p{}".format(engine._ns, rep, p), chain)
                if mine:
                    self.comm = comm
        assert self.comm is not None
        self.group_rank = {r: i for i, r in enumerate(self.stage_ranks)}
        self.schedule = engine.config.pipeline.strategy
        self.dtype = engine.dtype
        self.device = engine.device
        self._shape_cache = {}  # microbatch idx -> (shape, dtype)
        self._shapes_known = False

    # ---- mixed stage widths ---------------------------------------------------
    def _init_mixed(self, engine, rank_lists):
        """Stages with different device counts (e.g. a width-1 embedding
        stage feeding a dense-TP-2 stage).

        Only legal when every unequal boundary is 1<->k and the wide side
        is a ``replicate(..., replicated_io=True)`` stage: dense-TP blocks
        (ops/tp_mlp.py) keep boundary activations IDENTICAL across the
        stage's k positions (copy_to_group all-reduces input grads, the
        row-parallel all-reduce replicates outputs), so the boundary
        adapters are pure fan-out/fan-in of one tensor:

        * 1->k forward: the narrow rank sends the activation to all k
          positions (the Megatron 'copy' fan-out across the pipe).
        * 1->k backward: every position already holds the FULL summed
          input-grad after copy_to_group's backward all-reduce — the
          narrow rank receives from position 0 only (receiving k copies
          and summing would multiply the gradient by k).
        * k->1 forward: outputs are replicated — position 0 sends.
        * k->1 backward: the 'reduce' op's backward is identity per rank,
          so every position needs the same grad_out — the narrow rank
          sends it to all k positions.

        Expert-parallel stages carry per-position DATA STREAMS, not
        replicas, so they cannot change width; the replicated_io marker
        is how a stage declares itself safe (never guessed from module
        types).  Both schedules run: 1F1B's fused bidirectional
        exchanges decompose into per-link pairwise batches (the narrow
        rank batches with position 0; plain send/recv covers the other
        k-1 links), exercised by the mixed-width exact-match tests.
        """
        self.widths = [len(r) for r in rank_lists]
        S = self.S
        for s in range(S - 1):
            a, b = self.widths[s], self.widths[s + 1]
            if a != b:
                if 1 not in (a, b):
                    raise NotImplementedError(
                        "pipeline boundary stage_{} ({} ranks) -> stage_{}"
                        " ({} ranks): only 1<->k width changes are "
                        "supported".format(s, a, s + 1, b))
                wide = engine.stage_tgs[s if a > b else s + 1]
                if not wide.replicated_io:
                    raise ValueError(
                        "mixed-width pipeline boundary stage_{}({}) -> "
                        "stage_{}({}): the wide stage must be built with "
                        "epl.replicate(..., replicated_io=True) — only "
                        "stages whose boundary activations are identical "
                        "on every position (dense-TP blocks, ops/tp_mlp) "
                        "can change width; expert-parallel stages carry "
                        "per-position data streams and cannot".format(
                            s, a, s + 1, b))
        self.pos = rank_lists[self.s].index(engine.rank)
        w = self.widths
        self._next_is_narrow = self.s < S - 1 and w[self.s + 1] < w[self.s]
        self._prev_is_narrow = self.s > 0 and w[self.s - 1] < w[self.s]
        self._prev_is_wide = self.s > 0 and w[self.s - 1] > w[self.s]
        # per-boundary link communicators: (up_rank, down_rank) pairs.
        # gloo group creation is collective over ALL ranks — every rank
        # walks every (replica, boundary, link) in the same order.
        self._links_prev = []  # (comm, peer_group_rank, peer_global_rank)
        self._links_next = []
        for rep in range(engine.num_replicas):
            lists = [tg.virtual_device.local_ranks(rep)
                     for tg in engine.stage_tgs]
            for s in range(S - 1):
                up, dn = lists[s], lists[s + 1]
                if len(up) == len(dn):
                    pairs = list(zip(up, dn))
                elif len(up) == 1:
                    pairs = [(up[0], d) for d in dn]
                else:
                    pairs = [(u, dn[0]) for u in up]
                for li, (u, v) in enumerate(pairs):
                    mine = engine.rank in (u, v)
                    if not mine and (torch.cuda.is_available()
                                     or not dist.is_initialized()):
                        continue
                    comm = create_communicator(
                        "{}_pipeb_rep{}_s{}_l{}".format(
                            engine._ns, rep, s, li), [u, v])
                    if mine and rep == engine.replica_id:
                        if engine.rank == u:
                            self._links_next.append(
                                (comm, comm.ranks.index(v), v))
                        else:
                            self._links_prev.append(
                                (comm, comm.ranks.index(u), u))
        self.comm = None
        self.prev_rank = self._links_prev[0][2] if self._links_prev else None
        self.next_rank = self._links_next[0][2] if self._links_next else None
        self.schedule = engine.config.pipeline.strategy
        self.dtype = engine.dtype
        self.device = engine.device
        self._shape_cache = {}
        self._shapes_known = False

    # ---- shape handshake (first step only) -----------------------------------
    def _send_shape(self, t, peer):
        if self._shapes_known:
            return
        meta = [tuple(t.shape), str(t.dtype)]
        if dist.is_initialized():
            dist.send_object_list(meta, dst=peer,
                                  group=self.engine._control_group)

    def _recv_shape(self, i, peer):
        if i in self._shape_cache:
            return self._shape_cache[i]
        meta = [None, None]
        dist.recv_object_list(meta, src=peer,
                              group=self.engine._control_group)
        shape = tuple(meta[0])
        dtype = getattr(torch, meta[1].replace("torch.", ""))
        self._shape_cache[i] = (shape, dtype)
        return shape, dtype

    # ---- p2p primitives -------------------------------------------------------
    def _g(self, global_rank):
        return self.group_rank[global_rank]

    def recv_forward(self, i):
        if self.mixed:
            # narrow-after-wide receives from position 0 only; every other
            # case has exactly one upstream link
            comm, pg, peer = self._links_prev[0]
            shape, dtype = self._recv_shape(i, peer)
            t = torch.empty(shape, dtype=dtype, device=self.device)
            comm.recv(t, pg)
            return t
        shape, dtype = self._recv_shape(i, self.prev_rank)
        t = torch.empty(shape, dtype=dtype, device=self.device)
        self.comm.recv(t, self._g(self.prev_rank))
        return t

    def send_forward(self, t, i):
        if self.mixed:
            if self._next_is_narrow and self.pos != 0:
                return  # replicated output: position 0 alone sends
            t = t.contiguous()
            for comm, pg, peer in self._links_next:  # 1->k fans out
                self._send_shape(t, peer)
                comm.send(t, pg)
            return
        self._send_shape(t, self.next_rank)
        self.comm.send(t.contiguous(), self._g(self.next_rank))

    def recv_backward(self, i):
        shape, dtype = self._shape_cache_out[i]
        t = torch.empty(shape, dtype=dtype, device=self.device)
        if self.mixed:
            # narrow-before-wide takes the already-all-reduced grad from
            # position 0; wide positions each get their copy from narrow
            comm, pg, _ = self._links_next[0]
            comm.recv(t, pg)
            return t
        self.comm.recv(t, self._g(self.next_rank))
        return t

    def send_backward(self, g, i):
        if self.mixed:
            if self._prev_is_wide:  # k->1 backward: same grad to all k
                g = g.contiguous()
                for comm, pg, _ in self._links_prev:
                    comm.send(g, pg)
                return
            if self._prev_is_narrow and self.pos != 0:
                return  # grads already summed by copy_to_group's backward
            comm, pg, _ = self._links_prev[0]
            comm.send(g.contiguous(), pg)
            return
        self.comm.send(g.contiguous(), self._g(self.prev_rank))

    def send_forward_recv_backward(self, t, i_send, i_recv):
        shape, dtype = self._shape_cache_out[i_recv]
        g = torch.empty(shape, dtype=dtype, device=self.device)
        if self.mixed:
            # every pairwise link exchange is either a matching batched
            # send+recv (both ends batch on the same 2-rank comm) or a
            # matching plain send/recv — per-link pairing keeps the
            # bidirectional steady state deadlock-free at 1<->k
            # boundaries too
            if self._next_is_narrow and self.pos != 0:
                comm, pg, _ = self._links_next[0]
                comm.recv(g, pg)  # pure grad recv: position 0 sent fwd
                return g
            t = t.contiguous()
            comm0, pg0, peer0 = self._links_next[0]
            self._send_shape(t, peer0)
            comm0.batch_p2p([(True, t, pg0), (False, g, pg0)])
            for comm, pg, peer in self._links_next[1:]:  # 1->k fan-out
                self._send_shape(t, peer)
                comm.send(t, pg)
            return g
        self._send_shape(t, self.next_rank)
        self.comm.batch_p2p([
            (True, t.contiguous(), self._g(self.next_rank)),
            (False, g, self._g(self.next_rank)),
        ])
        return g

    def send_backward_recv_forward(self, g, i_recv):
        if self.mixed:
            if self._prev_is_narrow and self.pos != 0:
                # grads were summed by copy_to_group's backward; position
                # 0 alone sends — this position only receives forward
                return self.recv_forward(i_recv)
            g = g.contiguous()
            comm0, pg0, peer0 = self._links_prev[0]
            shape, dtype = self._recv_shape(i_recv, peer0)
            t = torch.empty(shape, dtype=dtype, device=self.device)
            comm0.batch_p2p([(True, g, pg0), (False, t, pg0)])
            if self._prev_is_wide:  # k->1: same grad to the other k-1
                for comm, pg, _ in self._links_prev[1:]:
                    comm.send(g, pg)
            return t
        shape, dtype = self._recv_shape(i_recv, self.prev_rank)
        t = torch.empty(shape, dtype=dtype, device=self.device)
        self.comm.batch_p2p([
            (True, g.contiguous(), self._g(self.prev_rank)),
            (False, t, self._g(self.prev_rank)),
        ])
        return t

    # ---- pipelined evaluation -------------------------------------------------
    def _send_eval(self, t):
        """Eval-time forward send with a per-call shape handshake (eval
        batches may differ from training shapes; eval is not hot)."""
        t = t.contiguous()
        targets = (self._links_next if self.mixed
                   else [(self.comm, self._g(self.next_rank),
                          self.next_rank)])
        if self.mixed and self._next_is_narrow and self.pos != 0:
            return  # replicated output: position 0 alone sends
        for comm, pg, peer in targets:
            meta = [tuple(t.shape), str(t.dtype)]
            dist.send_object_list(meta, dst=peer,
                                  group=self.engine._control_group)
            comm.send(t, pg)

    def _recv_eval(self):
        if self.mixed:
            comm, pg, peer = self._links_prev[0]
        else:
            comm, pg, peer = (self.comm, self._g(self.prev_rank),
                              self.prev_rank)
        meta = [None, None]
        dist.recv_object_list(meta, src=peer,
                              group=self.engine._control_group)
        shape = tuple(meta[0])
        dtype = getattr(torch, meta[1].replace("torch.", ""))
        t = torch.empty(shape, dtype=dtype, device=self.device)
        comm.recv(t, pg)
        return t

    def run_eval(self, inputs, num_micro_batch=None):
        """Forward-only pipelined evaluation (closes the reference's
        eval story under PP — hooks.py:915-933 has the chief evaluate
        while workers barrier; here the whole chain evaluates).

        COLLECTIVE across the replica's pipeline chain: every rank must
        call it with the same ``num_micro_batch`` (defaults to the
        training value).  ``inputs`` is consumed on stage 0; the LAST
        stage returns the concatenated outputs, every other stage
        returns None.  Runs in eval mode (dropout off, BatchNorm frozen)
        without grad.
        """
        M = num_micro_batch or self.engine.num_micro_batch
        mod = self.engine.stage_module
        was_training = mod.training
        mod.eval()
        outs = []
        try:
            with torch.no_grad():
                chunks = (torch.chunk(inputs, M, dim=0)
                          if self.s == 0 else [None] * M)
                if self.s == 0 and len(chunks) < M:
                    raise ValueError(
                        "eval batch dim {} < num_micro_batch {}".format(
                            inputs.shape[0], M))
                for i in range(M):
                    if self.s == 0:
                        x = chunks[i].to(self.device)
                    else:
                        x = self._recv_eval()
                    with self.engine.amp.autocast():
                        out = mod(x)
                    if self.s < self.S - 1:
                        self._send_eval(out)
                    else:
                        outs.append(out)
        finally:
            mod.train(was_training)
        return torch.cat(outs, dim=0) if outs else None

    # ---- forward/backward wrappers -------------------------------------------
    def _forward(self, inp, target):
        with self.engine.amp.autocast():
            out = self.engine.stage_module(inp)
            loss = None
            if self.s == self.S - 1:
                loss = self.engine.loss_fn(out, target)
        return out, loss

    def _backward(self, inp, out, loss, grad_out, is_last_backward):
        self.engine._set_reducers_enabled(is_last_backward)
        if loss is not None:
            self.engine.amp.scale_loss(loss).backward()
        else:
            out.backward(grad_out)
        if inp is not None and inp.requires_grad:
            g = inp.grad
            inp.grad = None
            return g
        return None

    # ---- schedules ------------------------------------------------------------
    def run(self, inputs, targets):
        M = self.engine.num_micro_batch
        input_chunks = (torch.chunk(inputs, M, dim=0)
                        if self.s == 0 else [None] * M)
        target_chunks = (torch.chunk(targets, M, dim=0)
                         if self.s == self.S - 1 else [None] * M)
        if self.s == 0 and len(input_chunks) < M:
            raise ValueError(
                "batch dim {} is smaller than pipeline.num_micro_batch={}"
                " — torch.chunk would silently produce fewer micro-"
                "batches".format(inputs.shape[0], M))
        self._shape_cache_out = {}
        self.engine._set_reducers_enabled(False)
        if self.schedule == constant.SCHEDULER_PREFER_FORWARD:
            loss = self._run_gpipe(M, input_chunks, target_chunks)
        else:
            loss = self._run_1f1b(M, input_chunks, target_chunks)
        self._shapes_known = True
        self.engine.finish_grad_sync()
        return loss

    def _get_input(self, i, input_chunks):
        if self.s == 0:
            x = input_chunks[i].to(self.device)
            return x
        x = self.recv_forward(i)
        x.requires_grad_(True)
        return x

    def _emit_forward(self, out, i):
        """Track output meta for matching backward recv."""
        self._shape_cache_out[i] = (tuple(out.shape), out.dtype)
        if self.s < self.S - 1:
            self.send_forward(out, i)

    def _run_gpipe(self, M, input_chunks, target_chunks):
        """PreferForward: all forwards, then all backwards
        (reference: scheduler.py:36-50)."""
        saved = []
        total_loss = None
        for i in range(M):
            x = self._get_input(i, input_chunks)
            out, loss = self._forward(x, target_chunks[i])
            self._emit_forward(out, i)
            saved.append((x, out, loss))
            if loss is not None:
                total_loss = loss.detach() if total_loss is None \
                    else total_loss + loss.detach()
        for i in range(M):
            x, out, loss = saved[i]
            grad_out = None
            if self.s < self.S - 1:
                grad_out = self.recv_backward(i)
            gin = self._backward(x, out, loss, grad_out, i == M - 1)
            if self.s > 0:
                self.send_backward(gin, i)
        return None if total_loss is None else total_loss / M

    def _run_1f1b(self, M, input_chunks, target_chunks):
        """PreferBackward (1F1B) — also used for PreferBackwardOptimizer
        (the apply overlap is inherent here: the fused optimizer launches
        right after the last backward, overlapping the tail of the DP
        allreduce).  Reference: scheduler.py:53-116."""
        warmup = min(self.S - 1 - self.s, M)
        remaining = M - warmup
        in_q = []
        total_loss = None

        def note_loss(loss):
            nonlocal total_loss
            if loss is not None:
                total_loss = (loss.detach() if total_loss is None
                              else total_loss + loss.detach())

        fi = 0  # forward microbatch index
        bi = 0  # backward microbatch index
        for _ in range(warmup):
            x = self._get_input(fi, input_chunks)
            out, loss = self._forward(x, target_chunks[fi])
            note_loss(loss)
            self._shape_cache_out[fi] = (tuple(out.shape), out.dtype)
            if self.s < self.S - 1:
                self.send_forward(out, fi)
            in_q.append((x, out, loss))
            fi += 1
        x = self._get_input(fi, input_chunks) if remaining > 0 else None
        for k in range(remaining):
            out, loss = self._forward(x, target_chunks[fi])
            note_loss(loss)
            self._shape_cache_out[fi] = (tuple(out.shape), out.dtype)
            if self.s < self.S - 1:
                grad_out = self.send_forward_recv_backward(out, fi, bi)
            else:
                grad_out = None
            in_q.append((x, out, loss))
            fi += 1
            ox, oout, oloss = in_q.pop(0)
            is_last = (k == remaining - 1) and warmup == 0
            gin = self._backward(ox, oout, oloss, grad_out, is_last)
            bi += 1
            if k != remaining - 1:
                if self.s > 0:
                    x = self.send_backward_recv_forward(gin, fi)
                    x.requires_grad_(True)
                else:
                    x = self._get_input(fi, input_chunks)
            else:
                if self.s > 0:
                    self.send_backward(gin, bi - 1)
                x = None
        for k in range(warmup):
            grad_out = (self.recv_backward(bi)
                        if self.s < self.S - 1 else None)
            ox, oout, oloss = in_q.pop(0)
            gin = self._backward(ox, oout, oloss, grad_out,
                                 k == warmup - 1)
            if self.s > 0:
                self.send_backward(gin, bi)
            bi += 1
        return None if total_loss is None else total_loss / M
