"""Cross-GPU KV-block migration over RCCL/xGMI.

The MI355X-native tier the reference lacks: any of the 8
GPUs on a node can pull a peer's cached blocks over point-to-point xGMI
links (7 links x ~153 GB/s per GPU) instead of recomputing or round-
tripping through host DRAM. Point-to-point send/recv is the right
primitive for one-to-one block pulls — ring collectives are per-link-bound
and the wrong shape.

Design: one process per GPU (torch.distributed), two dedicated process
groups (never shared with application collectives):

  - control plane (gloo): fixed-size int64 messages, a blocking any-source
    listener thread (PULL_REQ / PULL_ACK / BYE), sender rank embedded;
  - data plane (nccl=RCCL on GPU, gloo on CPU CI): isend/irecv of packed
    block slabs, issued ONLY from the single service thread
    (ProcessGroupNCCL is not thread-safe); packing/unpacking via the CDNA4
    gather/scatter kernels (BlockCopier) on a dedicated HIP stream.

Completion is backend-aware: NCCL works are event-polled
(is_completed()); gloo works get a waiter thread because gloo p2p
is_completed() never fires without wait(). The service loop never blocks
on a peer, so concurrent bidirectional pulls cannot deadlock.
"""
from __future__ import annotations

import concurrent.futures
import contextlib
import threading
import time
from collections import deque
from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Tuple

OP_PULL_REQ = 1
OP_PULL_ACK = 2
OP_BYE = 3
OP_PULL_REQ_MULTI = 4
OP_PULL_ACK_MULTI = 5

# Single-chunk ops use the first 8 slots
# [sender, op, req_id, group, chunk_hash, n_blocks, ok, pad]; batched ops
# put n_chunks in slot 3, a granted bitmask in slot 6 (ACK), and per-chunk
# (chunk_hash, group, n_blocks) triples from slot 8. All control messages
# are the same fixed size so the any-source listener recv stays simple.
MAX_BATCH = 16
MSG_LEN = 8 + 3 * MAX_BATCH


@dataclass
class PeerStats:
    pulls_requested: int = 0
    pulls_served: int = 0
    pulls_served_dram: int = 0
    pulls_failed: int = 0
    bytes_sent: int = 0
    bytes_received: int = 0


class _OpTracker:
    """Uniform completion check over gloo (waiter thread on wait()) and
    NCCL (event polling via is_completed())."""

    def __init__(self, work, use_wait_thread: bool):
        self.work = work
        self._ev: Optional[threading.Event] = None
        if use_wait_thread:
            self._ev = threading.Event()
            threading.Thread(target=self._waiter, daemon=True).start()

    def _waiter(self):
        try:
            self.work.wait()
        except Exception:
            pass
        finally:
            self._ev.set()

    def done(self) -> bool:
        if self._ev is not None:
            return self._ev.is_set()
        return self.work.is_completed()


class PeerMigrationService:
    def __init__(self, groups: Sequence[Sequence], data_group=None,
                 control_group=None, device: Optional[int] = None,
                 poll_interval_s: float = 0.0002, dram_lookup=None):
        """groups: per KV-cache group, a list of per-layer page tensors
        (the offload engine's layout). data_group: nccl(=RCCL) on GPU /
        gloo on CPU. control_group: gloo. Both must be dedicated groups.
        dram_lookup: optional (chunk_hash, group, n_blocks) ->
        (host_uint8_tensor, is_fp8) | None — lets this rank serve pulls
        from its pinned host-DRAM cache after HBM eviction (see
        peer.tiered.make_dram_lookup)."""
        import torch
        import torch.distributed as dist

        from .. import ensure_offload_native

        _kvoffload = ensure_offload_native()

        self._torch = torch
        self._dist = dist
        self.data_group = data_group
        self.control_group = control_group
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        self.gpu_mode = groups[0][0].is_cuda
        self.device = device if device is not None else (
            groups[0][0].device.index or 0 if self.gpu_mode else 0)
        self._poll = poll_interval_s
        self._dram_lookup = dram_lookup
        self._data_needs_wait_thread = not self.gpu_mode  # gloo data plane

        native_groups = []
        self._geo = []
        for g in groups:
            ptrs = [t.data_ptr() for t in g]
            strides = [t.stride(0) * t.element_size() for t in g]
            native_groups.append((ptrs, strides, strides[0], int(g[0].shape[0])))
            self._geo.append({"num_layers": len(g), "block_bytes": strides[0]})
        self._copier = _kvoffload.BlockCopier(native_groups, self.gpu_mode,
                                              self.device)
        self._tensors = [list(g) for g in groups]

        self._registry: Dict[Tuple[int, int], List[int]] = {}
        self._reg_mu = threading.Lock()
        self._cmd_q: deque = deque()
        self._ctrl_q: deque = deque()
        self._q_mu = threading.Lock()
        self._stats = PeerStats()
        self._abandoned: List[tuple] = []  # timed-out (tracker, buf) pairs
        self._stopping = False
        self._byes_seen = 0
        self._next_req_id = self.rank + 1
        if self.gpu_mode:
            self._comm_stream = torch.cuda.Stream(device=self.device)
        self._listener = threading.Thread(target=self._ctrl_listen, daemon=True,
                                          name=f"peer-ctrl-{self.rank}")
        self._listener.start()
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name=f"peer-svc-{self.rank}")
        self._thread.start()

    # ---- registry -----------------------------------------------------------

    def register_blocks(self, chunk_hash: int, group: int,
                        block_ids: Sequence[int]) -> None:
        with self._reg_mu:
            self._registry[(chunk_hash, group)] = [int(b) for b in block_ids]

    def unregister_blocks(self, chunk_hash: int, group: int) -> None:
        with self._reg_mu:
            self._registry.pop((chunk_hash, group), None)

    def lookup_local(self, chunk_hash: int, group: int):
        with self._reg_mu:
            return self._registry.get((chunk_hash, group))

    def stats(self) -> PeerStats:
        return self._stats

    # ---- API ----------------------------------------------------------------

    def pull(self, chunk_hash: int, group: int, dst_block_ids: Sequence[int],
             src_rank: int, timeout: float = 30.0,
             fp8: bool = False) -> concurrent.futures.Future:
        """Pull a peer's cached chunk into local pages. The Future resolves
        to True (pulled), False (peer does not hold it), or raises on
        timeout/shutdown. fp8=True asks the peer to quantize HBM-resident
        blocks to fp8 e4m3 on the fly — half the xGMI bytes, e4m3 rounding
        on the payload (DRAM-cached chunks always travel in their stored
        codec; the ACK says which arrived)."""
        ids = [int(b) for b in dst_block_ids]
        if not ids or len(ids) > 64:
            raise ValueError("dst_block_ids must have 1..64 blocks")
        fut: concurrent.futures.Future = concurrent.futures.Future()
        with self._q_mu:
            self._cmd_q.append(("pull", chunk_hash, group, ids, src_rank,
                                time.time() + timeout, fut, bool(fp8)))
        return fut

    def pull_many(self, chunks: Sequence[Tuple[int, int, Sequence[int]]],
                  src_rank: int, timeout: float = 30.0,
                  fp8: bool = False) -> concurrent.futures.Future:
        """Batched pull: up to MAX_BATCH (chunk_hash, group, dst_block_ids)
        from ONE peer with one control round trip and one xGMI data
        transfer (small chunks amortize the handshake and the per-transfer
        launch cost). The Future resolves to a list[bool] per chunk (True =
        pulled); raises on timeout/shutdown."""
        if not chunks or len(chunks) > MAX_BATCH:
            raise ValueError(f"pull_many takes 1..{MAX_BATCH} chunks")
        fut: concurrent.futures.Future = concurrent.futures.Future()
        norm = [(int(h), int(g), [int(b) for b in ids])
                for (h, g, ids) in chunks]
        for _, _, ids in norm:
            if not ids or len(ids) > 64:
                raise ValueError("each chunk must have 1..64 blocks")
        with self._q_mu:
            self._cmd_q.append(("pull_many", norm, src_rank,
                                time.time() + timeout, fut, bool(fp8)))
        return fut

    def close(self) -> None:
        """Collective: every rank must call close() (the shutdown protocol
        synchronizes on the control group so no listener is left blocked
        in a recv — a pending gloo recv at process-group destruction
        aborts the process)."""
        if self._stopping:
            return
        # Phase 1: stop the listener-exit flag BEFORE any BYE can arrive.
        self._stopping = True
        with self._q_mu:
            self._cmd_q.append(("bye",))
        self._thread.join(timeout=30.0)
        # The LISTENER must be joined too: it exits only after draining
        # every peer's BYE, and destroying (or GC-ing at interpreter exit)
        # the control group while it still sits in dist.recv aborts the
        # whole process inside gloo ("terminate called without an active
        # exception" — reproduced 10/12 at world 8 with an explicit
        # destroy_process_group racing the listener).
        self._listener.join(timeout=30.0)

    # ---- control listener (blocking gloo recv, any source) ------------------

    def _ctrl_listen(self) -> None:
        torch, dist = self._torch, self._dist
        while True:
            buf = torch.zeros(MSG_LEN, dtype=torch.int64)
            try:
                dist.recv(buf, src=None, group=self.control_group)
            except Exception:
                return  # process group torn down
            if int(buf[1]) == OP_BYE:
                self._byes_seen += 1
            else:
                with self._q_mu:
                    self._ctrl_q.append(buf)
            # Exit only after draining every peer's BYE: unconsumed gloo
            # buffers (or a still-pending recv) at process-group
            # destruction abort the process. The close() barrier
            # guarantees exactly world-1 BYEs arrive after _stopping.
            if self._stopping and self._byes_seen >= self.world - 1:
                return

    # ---- service loop -------------------------------------------------------

    def _msg(self, op, req_id=0, group=0, chunk_hash=0, n_blocks=0, ok=0):
        t = self._torch.zeros(MSG_LEN, dtype=self._torch.int64)
        u = chunk_hash & ((1 << 64) - 1)
        t[0], t[1], t[2], t[3] = self.rank, op, req_id, group
        t[4] = u - (1 << 64) if u >= (1 << 63) else u
        t[5], t[6] = n_blocks, ok
        return t

    def _data_tensor(self, group: int, n_blocks: int):
        nb = self._copier.packed_bytes(group, n_blocks)
        if self.gpu_mode:
            return self._torch.empty(nb, dtype=self._torch.uint8, device="cuda")
        return self._torch.empty(nb, dtype=self._torch.uint8)

    def _track(self, work) -> _OpTracker:
        return _OpTracker(work, self._data_needs_wait_thread)

    def _loop(self) -> None:
        torch, dist = self._torch, self._dist
        if self.gpu_mode:
            torch.cuda.set_device(self.device)
        pending_pulls: Dict[int, dict] = {}
        pending_sends: List[Tuple[_OpTracker, object]] = []
        ctx = (torch.cuda.stream(self._comm_stream) if self.gpu_mode
               else contextlib.nullcontext())
        with ctx:
            run = True
            while run or pending_sends:
                made_progress = False

                item = None
                with self._q_mu:
                    if self._ctrl_q:
                        item = ("ctrl", self._ctrl_q.popleft())
                    elif run and self._cmd_q:
                        item = ("cmd", self._cmd_q.popleft())
                if item is not None:
                    made_progress = True
                    kind, payload = item
                    try:
                        if kind == "cmd":
                            if payload[0] == "bye":
                                # Two-phase shutdown: barrier first so every
                                # rank's _stopping is set before ANY BYE is
                                # sent; each listener's next recv then
                                # observes _stopping and exits. Without
                                # this, a listener whose peers closed early
                                # blocks in recv forever and the pending op
                                # aborts process-group destruction.
                                with contextlib.suppress(Exception):
                                    dist.barrier(group=self.control_group)
                                for p in range(self.world):
                                    if p != self.rank:
                                        with contextlib.suppress(Exception):
                                            dist.send(self._msg(OP_BYE), dst=p,
                                                      group=self.control_group)
                                run = False
                            elif payload[0] == "pull_many":
                                self._start_pull_many(payload, pending_pulls)
                            else:
                                self._start_pull(payload, pending_pulls)
                        else:
                            self._handle_ctrl(payload, pending_pulls,
                                              pending_sends)
                    except Exception as e:
                        # a malformed request must not kill the service; the
                        # affected pull fails by timeout on the requester
                        import logging

                        logging.getLogger(__name__).error(
                            "peer service error handling %s: %s", kind, e)
                        if kind == "cmd" and payload[0] != "bye":
                            # pull payloads end with (..., fut, want_fp8)
                            fut = payload[-2]
                            if not fut.done():
                                fut.set_exception(e)

                for req_id in list(pending_pulls):
                    st = pending_pulls[req_id]
                    if st["tracker"] is not None and st["tracker"].done():
                        made_progress = True
                        try:
                            self._finish_pull(st)
                        except Exception as e:
                            st["fut"].set_exception(e)
                            self._stats.pulls_failed += 1
                        del pending_pulls[req_id]
                    elif time.time() > st["deadline"]:
                        st["fut"].set_exception(
                            TimeoutError(f"pull {req_id} from rank {st['src']}"))
                        self._stats.pulls_failed += 1
                        # the posted data irecv cannot be cancelled: keep the
                        # tracker AND the buffer alive for the service
                        # lifetime, or a late-arriving send would complete
                        # into freed memory
                        if st.get("tracker") is not None:
                            self._abandoned.append((st["tracker"], st["buf"]))
                        del pending_pulls[req_id]
                before = len(pending_sends)
                pending_sends[:] = [(t, b) for (t, b) in pending_sends
                                    if not t.done()]
                made_progress |= len(pending_sends) != before

                if not made_progress:
                    if not run and not pending_pulls and not pending_sends:
                        break
                    time.sleep(self._poll)

        for st in pending_pulls.values():
            if not st["fut"].done():
                st["fut"].set_exception(RuntimeError("peer service closed"))

    def _finish_pull(self, st) -> None:
        """Scatter a completed transfer into the destination pages and
        resolve the future (exceptions surface on the future, never kill
        the service loop)."""
        stream = self._comm_stream.cuda_stream if self.gpu_mode else 0
        if "chunks" in st:  # batched pull
            base = st["buf"].data_ptr()
            off = 0
            for ci, (h, g, ids) in enumerate(st["chunks"]):
                if not st["granted"][ci]:
                    continue
                if st["fp8_chunks"][ci]:
                    self._copier.scatter_fp8(g, ids, base + off, stream)
                    off += self._copier.packed_bytes_fp8(g, len(ids))
                else:
                    self._copier.scatter(g, ids, base + off, stream)
                    off += self._copier.packed_bytes(g, len(ids))
            result = st["granted"]
        else:
            if st.get("fp8"):
                self._copier.scatter_fp8(st["group"], st["dst_ids"],
                                         st["buf"].data_ptr(), stream)
            else:
                self._copier.scatter(st["group"], st["dst_ids"],
                                     st["buf"].data_ptr(), stream)
            result = True
        if self.gpu_mode:
            self._comm_stream.synchronize()
        self._stats.bytes_received += st["buf"].numel()
        st["fut"].set_result(result)

    def _local_serve(self, chunk_hash, group, dst_ids):
        """Self-pull short circuit (RCCL cannot send to self): resolve from
        local HBM registry or the DRAM cache straight into the dst pages."""
        stream = self._comm_stream.cuda_stream if self.gpu_mode else 0
        ids = self.lookup_local(chunk_hash, group)
        if ids is not None and len(ids) == len(dst_ids):
            slab = self._data_tensor(group, len(dst_ids))
            self._copier.gather(group, ids, slab.data_ptr(), stream)
            self._copier.scatter(group, dst_ids, slab.data_ptr(), stream)
            self._stats.pulls_served += 1
            return True
        if self._dram_lookup is not None:
            dram = self._serve_dram(chunk_hash, group, len(dst_ids))
            if dram is not None:
                host, fp8 = dram
                buf = host.cuda() if self.gpu_mode else host
                if fp8:
                    self._copier.scatter_fp8(group, dst_ids, buf.data_ptr(),
                                             stream)
                else:
                    self._copier.scatter(group, dst_ids, buf.data_ptr(),
                                         stream)
                self._stats.pulls_served += 1
                self._stats.pulls_served_dram += 1
                return True
        return False

    def _start_pull(self, payload, pending_pulls) -> None:
        _, chunk_hash, group, dst_ids, src, deadline, fut, want_fp8 = payload
        if src == self.rank:  # local: no wire, fp8 request is moot
            ok = self._local_serve(chunk_hash, group, dst_ids)
            if self.gpu_mode:
                self._comm_stream.synchronize()
            if not ok:
                self._stats.pulls_failed += 1
            self._stats.pulls_requested += 1
            fut.set_result(ok)
            return
        req_id = self._next_req_id
        self._next_req_id += self.world
        t = self._msg(OP_PULL_REQ, req_id, group, chunk_hash, len(dst_ids))
        t[7] = 1 if want_fp8 else 0
        self._dist.send(t, dst=src, group=self.control_group)
        pending_pulls[req_id] = {
            "dst_ids": dst_ids, "group": group, "src": src,
            "deadline": deadline, "fut": fut, "tracker": None, "buf": None,
        }
        self._stats.pulls_requested += 1

    def _gather_fp8_slab(self, group, ids, stream):
        """Quantizing gather of HBM blocks into an exact-size fp8 slab."""
        nb = self._copier.packed_bytes_fp8(group, len(ids))
        if self.gpu_mode:
            scratch_b = self._copier.fp8_scratch_bytes(group, len(ids))
            scratch = self._torch.empty(scratch_b, dtype=self._torch.uint8,
                                        device="cuda")
            slab = self._torch.empty(nb, dtype=self._torch.uint8,
                                     device="cuda")
            self._copier.gather_fp8(group, ids, slab.data_ptr(),
                                    scratch.data_ptr(), stream)
        else:
            slab = self._torch.empty(nb, dtype=self._torch.uint8)
            self._copier.gather_fp8(group, ids, slab.data_ptr(), 0, 0)
        return slab

    def _serve_dram(self, chunk_hash, group, n_blocks):
        """DRAM-tier lookup with error isolation: a broken callback must
        not kill the service loop."""
        try:
            got = self._dram_lookup(chunk_hash, group, n_blocks)
        except Exception as e:
            import logging

            logging.getLogger(__name__).warning("dram_lookup failed: %s", e)
            return None
        if got is None:
            return None
        host, fp8 = got
        want = (self._copier.packed_bytes_fp8(group, n_blocks) if fp8
                else self._copier.packed_bytes(group, n_blocks))
        if host.numel() < want:
            return None
        return host[:want].contiguous(), fp8

    def _start_pull_many(self, payload, pending_pulls) -> None:
        _, chunks, src, deadline, fut, want_fp8 = payload
        if src == self.rank:  # local: no wire, fp8 request is moot
            res = [self._local_serve(h, g, ids) for (h, g, ids) in chunks]
            if self.gpu_mode:
                self._comm_stream.synchronize()
            self._stats.pulls_requested += len(chunks)
            self._stats.pulls_failed += sum(1 for r in res if not r)
            fut.set_result(res)
            return
        req_id = self._next_req_id
        self._next_req_id += self.world
        t = self._msg(OP_PULL_REQ_MULTI, req_id)
        t[3] = len(chunks)
        t[7] = 1 if want_fp8 else 0
        for i, (h, g, ids) in enumerate(chunks):
            u = h & ((1 << 64) - 1)
            t[8 + 3 * i] = u - (1 << 64) if u >= (1 << 63) else u
            t[9 + 3 * i] = g
            t[10 + 3 * i] = len(ids)
        self._dist.send(t, dst=src, group=self.control_group)
        pending_pulls[req_id] = {
            "chunks": chunks, "src": src, "deadline": deadline, "fut": fut,
            "tracker": None, "buf": None, "granted": None,
        }
        self._stats.pulls_requested += len(chunks)

    def _handle_ctrl(self, buf, pending_pulls, pending_sends) -> None:
        dist = self._dist
        sender = int(buf[0])
        op = int(buf[1])
        req_id = int(buf[2])
        group = int(buf[3])
        chunk_hash = int(buf[4]) & ((1 << 64) - 1)
        n_blocks = int(buf[5])
        ok = int(buf[6])
        if op == OP_PULL_REQ:
            want_fp8 = bool(int(buf[7]))
            ids = self.lookup_local(chunk_hash, group)
            grant = ids is not None and len(ids) == n_blocks
            dram = None
            if not grant and self._dram_lookup is not None:
                dram = self._serve_dram(chunk_hash, group, n_blocks)
            if grant:
                ok_code = 2 if want_fp8 else 1
            else:
                ok_code = (0 if dram is None else (2 if dram[1] else 1))
            dist.send(self._msg(OP_PULL_ACK, req_id, group, chunk_hash,
                                n_blocks, ok_code),
                      dst=sender, group=self.control_group)
            if grant:
                stream = (self._comm_stream.cuda_stream
                          if self.gpu_mode else 0)
                if want_fp8:
                    slab = self._gather_fp8_slab(group, ids, stream)
                else:
                    slab = self._data_tensor(group, n_blocks)
                    self._copier.gather(group, ids, slab.data_ptr(), stream)
                w = dist.isend(slab, dst=sender, group=self.data_group)
                pending_sends.append((self._track(w), slab))
                self._stats.pulls_served += 1
                self._stats.bytes_sent += slab.numel()
            elif dram is not None:
                slab = dram[0].cuda() if self.gpu_mode else dram[0]
                w = dist.isend(slab, dst=sender, group=self.data_group)
                pending_sends.append((self._track(w), slab))
                self._stats.pulls_served += 1
                self._stats.pulls_served_dram += 1
                self._stats.bytes_sent += slab.numel()
        elif op == OP_PULL_ACK:
            st = pending_pulls.get(req_id)
            if st is None:
                return
            if ok:
                st["fp8"] = ok == 2
                if st["fp8"]:
                    nb = self._copier.packed_bytes_fp8(st["group"],
                                                       len(st["dst_ids"]))
                    slab = (self._torch.empty(nb, dtype=self._torch.uint8,
                                              device="cuda")
                            if self.gpu_mode else
                            self._torch.empty(nb, dtype=self._torch.uint8))
                else:
                    slab = self._data_tensor(st["group"], len(st["dst_ids"]))
                st["buf"] = slab
                w = dist.irecv(slab, src=st["src"], group=self.data_group)
                st["tracker"] = self._track(w)
            else:
                st["fut"].set_result(False)
                self._stats.pulls_failed += 1
                del pending_pulls[req_id]
        elif op == OP_PULL_REQ_MULTI:
            want_fp8 = bool(int(buf[7]))
            n_chunks = int(buf[3])
            reqs = []
            for i in range(n_chunks):
                h = int(buf[8 + 3 * i]) & ((1 << 64) - 1)
                g = int(buf[9 + 3 * i])
                nb = int(buf[10 + 3 * i])
                reqs.append((h, g, nb))
            mask = 0
            fp8_mask = 0
            total = 0
            served = []  # ("hbm", g, ids) | ("dram", host_tensor)
            n_dram = 0
            for i, (h, g, nb) in enumerate(reqs):
                ids = self.lookup_local(h, g)
                if ids is not None and len(ids) == nb:
                    mask |= 1 << i
                    if want_fp8:
                        fp8_mask |= 1 << i
                        total += self._copier.packed_bytes_fp8(g, nb)
                    else:
                        total += self._copier.packed_bytes(g, nb)
                    served.append(("hbm", g, ids))
                    continue
                if self._dram_lookup is not None:
                    dram = self._serve_dram(h, g, nb)
                    if dram is not None:
                        mask |= 1 << i
                        if dram[1]:
                            fp8_mask |= 1 << i
                        total += dram[0].numel()
                        served.append(("dram", dram[0], None))
                        n_dram += 1
            ack = self._msg(OP_PULL_ACK_MULTI, req_id)
            ack[3] = n_chunks
            ack[6] = mask
            ack[7] = fp8_mask
            dist.send(ack, dst=sender, group=self.control_group)
            if mask:
                slab = (self._torch.empty(total, dtype=self._torch.uint8,
                                          device="cuda") if self.gpu_mode
                        else self._torch.empty(total,
                                               dtype=self._torch.uint8))
                off = 0
                stream = (self._comm_stream.cuda_stream
                          if self.gpu_mode else 0)
                for item in served:
                    if item[0] == "hbm":
                        _, g, ids = item
                        if want_fp8:
                            part = self._gather_fp8_slab(g, ids, stream)
                            slab[off:off + part.numel()].copy_(part)
                            off += part.numel()
                        else:
                            self._copier.gather(g, ids,
                                                slab.data_ptr() + off,
                                                stream)
                            off += self._copier.packed_bytes(g, len(ids))
                    else:
                        host = item[1]
                        slab[off:off + host.numel()].copy_(host)
                        off += host.numel()
                w = dist.isend(slab, dst=sender, group=self.data_group)
                pending_sends.append((self._track(w), slab))
                self._stats.pulls_served += len(served)
                self._stats.pulls_served_dram += n_dram
                self._stats.bytes_sent += slab.numel()
        elif op == OP_PULL_ACK_MULTI:
            st = pending_pulls.get(req_id)
            if st is None:
                return
            mask = ok
            fp8_mask = int(buf[7])
            granted = [bool(mask >> i & 1) for i in range(len(st["chunks"]))]
            fp8_chunks = [bool(fp8_mask >> i & 1)
                          for i in range(len(st["chunks"]))]
            st["granted"] = granted
            st["fp8_chunks"] = fp8_chunks
            if mask:
                total = sum(
                    (self._copier.packed_bytes_fp8(g, len(ids))
                     if fp8_chunks[gi] else
                     self._copier.packed_bytes(g, len(ids)))
                    for gi, (h, g, ids) in enumerate(st["chunks"])
                    if granted[gi])
                slab = (self._torch.empty(total, dtype=self._torch.uint8,
                                          device="cuda") if self.gpu_mode
                        else self._torch.empty(total,
                                               dtype=self._torch.uint8))
                st["buf"] = slab
                w = dist.irecv(slab, src=st["src"], group=self.data_group)
                st["tracker"] = self._track(w)
            else:
                st["fut"].set_result([False] * len(st["chunks"]))
                self._stats.pulls_failed += len(st["chunks"])
                del pending_pulls[req_id]
