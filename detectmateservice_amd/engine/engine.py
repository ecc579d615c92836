"""Engine: the batched recv → process → send loop.

Reference behavior being reproduced (/root/reference/src/service/features/
engine.py:73-342): a daemon thread listens on ``engine_addr``, polls with a
recv timeout (100 ms default), hands frames to the processor, skips
``None`` results (filtered messages, engine.py:238-240), broadcasts results
to every ``out_addr`` socket or — when no outputs are configured — replies
on the input pair socket (request/reply compatibility, engine.py:248-264).
Sends are non-blocking with ``engine_retry_count`` retries at 10 ms and
drop-with-counter on failure (engine.py:281-301). Output sockets dial
non-blocking with background reconnect (engine.py:173-175). ``stop()``
joins the thread with a 2 s deadline and raises ``EngineException`` if it
fails to stop cleanly (engine.py:317-323); the thread is re-created on
restart (engine.py:185-191). Setup failure closes the input socket to
avoid leaks (engine.py:122-129).

MI355X-first departure: the loop is **batch-first** —
``recv_many`` drains up to ``engine_batch_size`` frames (bounded by
``engine_batch_linger_ms``) and calls ``processor.process_batch(frames)``
once, so a GPU component launches one kernel per batch instead of one
Python call per message. Per-frame semantics (ordering, None-skip,
per-output drop accounting) are preserved exactly.
"""
from __future__ import annotations

import logging
import os
import threading
import time
from typing import List, Optional, Protocol, runtime_checkable

from ..settings import ServiceSettings
from ..utils.metrics import ServiceMetrics
from .sockets import PairSocketFactory, RecvTimeout, SocketClosed


class EngineException(Exception):
    pass


@runtime_checkable
class Processor(Protocol):
    """Processing contract (reference engine.py:61-69) + batched extension."""

    def process(self, data: bytes) -> Optional[bytes]: ...

    def process_batch(self, frames: List[bytes]) -> List[Optional[bytes]]: ...


def _count_lines(data: bytes) -> int:
    """One frame = one log line; embedded newlines add more (reference
    counts ``raw.count(b'\\n')`` — engine.py:213 — which is 0 for protobuf
    frames; we floor at 1 so lines/sec matches frames of protobuf traffic)."""
    n = data.count(b"\n")
    return n if n > 0 else 1


class Engine:
    def __init__(
        self,
        settings: ServiceSettings,
        processor: Processor,
        socket_factory: Optional[PairSocketFactory] = None,
        logger: Optional[logging.Logger] = None,
        metrics: Optional[ServiceMetrics] = None,
        dist_ctx=None,
    ) -> None:
        self.settings = settings
        self.processor = processor
        #: settings-driven placement (core.py builds it from dist_mode):
        #: an object with .mode/.rank/.world/.src — selects the fanout /
        #: stage loops below instead of the socket loop
        self._dist_ctx = dist_ctx
        self._log = logger or logging.getLogger(__name__)
        self._factory = socket_factory or PairSocketFactory()
        self.metrics = metrics or ServiceMetrics(
            settings.component_type, settings.component_id or "unknown"
        )

        self._stop_event = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._running = False

        # Bind the input socket now (reference engine.py:111-117); on any
        # failure during output setup, close the input to avoid leaks
        # (engine.py:122-129).
        self._pair_sock = self._factory.create(
            settings.engine_addr,
            logger=self._log,
            tls_config=settings.tls_input,
            buffer_size=settings.engine_buffer_size,
        )
        try:
            self._out_socks = self._setup_output_sockets()
        except Exception:
            self._pair_sock.close()
            raise

    # ------------------------------------------------------------------
    def _setup_output_sockets(self):
        socks = []
        for addr in self.settings.out_addr:
            kwargs = dict(
                logger=self._log,
                tls_config=self.settings.tls_output,
                buffer_size=self.settings.engine_buffer_size,
            )
            try:
                socks.append(self._factory.create_dialer(
                    addr, dial_timeout_s=self.settings.dial_timeout / 1000.0,
                    **kwargs,
                ))
            except TypeError:
                # injected test factories may not take dial_timeout_s
                socks.append(self._factory.create_dialer(addr, **kwargs))
        return socks

    @property
    def running(self) -> bool:
        return self._running and self._thread is not None and self._thread.is_alive()

    def start(self) -> None:
        if self.running:
            self._log.debug("engine already running")
            return
        # Thread is re-created on every start (reference engine.py:185-191).
        self._stop_event.clear()
        self._running = True
        self._thread = threading.Thread(
            target=self._run_loop, name="EngineLoop", daemon=True
        )
        self._thread.start()
        self.metrics.engine_starts_total.inc()
        self.metrics.engine_running.state("running")

    def stop(self) -> None:
        self._stop_event.set()
        self._running = False
        thread = self._thread
        if thread is not None and thread.is_alive():
            thread.join(timeout=2.0)
            if thread.is_alive():
                raise EngineException("Engine thread failed to stop cleanly")
        self.metrics.engine_running.state("stopped")

    def close(self) -> None:
        """Close all sockets (after stop)."""
        try:
            self._pair_sock.close()
        except Exception:  # noqa: BLE001
            pass
        for s in self._out_socks:
            try:
                s.close()
            except Exception:  # noqa: BLE001
                pass

    # ------------------------------------------------------------------
    def _run_loop(self) -> None:
        ctx = self._dist_ctx
        if ctx is not None and ctx.world > 1 and ctx.mode in ("fanout", "stage"):
            return self._run_dist_loop()
        if self.settings.engine_source_mode:
            return self._run_source_loop()
        if self.settings.engine_packed_mode and self._try_packed_loop():
            return
        return self._run_socket_loop()

    def _run_socket_loop(self) -> None:
        """The plain batched frame loop (also the elastic-degradation
        target when a dist collective path loses a peer)."""
        s = self.settings
        m = self.metrics
        self._log.info(
            "engine loop started on %s (batch_size=%d linger=%.1fms outputs=%d)",
            s.engine_addr, s.engine_batch_size, s.engine_batch_linger_ms,
            len(self._out_socks),
        )
        # DMX_ENGINE_STATS=1: per-batch stage timing every ~2 s (perf triage)
        stats_on = os.environ.get("DMX_ENGINE_STATS") == "1"
        st = {"n": 0, "frames": 0, "recv": 0.0, "proc": 0.0, "other": 0.0,
              "last": time.perf_counter()}
        t_end = time.perf_counter()
        while not self._stop_event.is_set():
            try:
                t_r0 = time.perf_counter()
                frames = self._pair_sock.recv_many(
                    s.engine_batch_size, s.engine_recv_timeout, s.engine_batch_linger_ms
                )
                if stats_on:
                    st["recv"] += time.perf_counter() - t_r0
            except RecvTimeout:
                continue
            except SocketClosed:
                break
            except OSError as exc:
                if self._stop_event.is_set():
                    break
                self._log.error("recv error: %s", exc)
                continue
            if not frames:
                continue
            # Skip empty frames (reference engine.py:207-209) — but keep
            # each surviving frame's ORIGINAL batch index: the listener's
            # _batch_conns was recorded for the unfiltered recv_many batch,
            # so reply(i) must be given pre-filter indices or multi-peer
            # replies route to the wrong sender.
            orig_idx = [i for i, f in enumerate(frames) if f]
            if len(orig_idx) != len(frames):
                frames = [frames[i] for i in orig_idx]
            if not frames:
                continue

            read_bytes = sum(len(f) for f in frames)
            read_lines = sum(_count_lines(f) for f in frames)
            m.data_read_bytes_total.inc(read_bytes)
            m.data_read_lines_total.inc(read_lines)
            m.engine_batch_size.observe(len(frames))

            t0 = time.perf_counter()
            try:
                outs = self.processor.process_batch(frames)
            except Exception as exc:  # noqa: BLE001 - loop must survive (engine.py:233-236)
                m.processing_errors_total.inc(len(frames))
                self._log.error("processing error on batch of %d: %s", len(frames), exc)
                continue
            elapsed = time.perf_counter() - t0
            m.data_processed_bytes_total.inc(read_bytes)
            m.data_processed_lines_total.inc(read_lines)
            m.observe_batch(elapsed, read_lines)

            for i, out in enumerate(outs):
                if out is None:
                    continue  # filtered (engine.py:238-240)
                if self._out_socks:
                    self._send_to_outputs(out)
                else:
                    # request/reply fallback mode (engine.py:248-264);
                    # replies route to EACH frame's sender (multi-peer)
                    reply = getattr(self._pair_sock, "reply", None)
                    ok = (reply(orig_idx[i], out) if reply is not None
                          else self._pair_sock.send(out, block=False))
                    if ok:
                        m.data_written_bytes_total.inc(len(out))
                        m.data_written_lines_total.inc(_count_lines(out))
                    else:
                        m.data_dropped_bytes_total.inc(len(out))
                        m.data_dropped_lines_total.inc(_count_lines(out))
            if stats_on:
                now = time.perf_counter()
                st["n"] += 1
                st["frames"] += len(frames)
                st["proc"] += elapsed
                st["other"] += (now - t_end) if t_end else 0.0
                t_end = now
                if now - st["last"] > 2.0 and st["n"]:
                    tot = st["other"]
                    self._log.info(
                        "[stats] %d batches (%.0f fr/batch): recv-wait %.1fms/b, "
                        "process %.1fms/b, other %.1fms/b, %.0f lines/s",
                        st["n"], st["frames"] / st["n"],
                        st["recv"] * 1e3 / st["n"], st["proc"] * 1e3 / st["n"],
                        (tot - st["recv"] - st["proc"]) * 1e3 / st["n"],
                        st["frames"] / tot if tot > 0 else 0.0,
                    )
                    st.update(n=0, frames=0, recv=0.0, proc=0.0, other=0.0,
                              last=now)
        self._log.info("engine loop exited")

    def _try_packed_loop(self) -> bool:
        """Native packed data plane: reader threads decode LogSchema frames
        straight into tensors (frame_reader.cpp read_batch_packed); the
        loop hands (lines, lens, ids) to the component with ZERO per-frame
        Python objects. Requires a plain fd listener, the C++ extension
        and a component exposing ``process_packed_frames``. Returns False
        (caller falls back to the frame loop) when unavailable."""
        s = self.settings
        m = self.metrics
        proc = getattr(self.processor, "process_packed_frames", None)
        supports = getattr(self.processor, "supports_packed_frames", None)
        if proc is None or (supports is not None and not supports()):
            self._log.warning(
                "engine_packed_mode: component has no process_packed_frames; "
                "using the frame loop"
            )
            return False
        enable = getattr(self._pair_sock, "enable_packed", None)
        max_len = getattr(self.processor, "packed_max_len", lambda: 256)()
        pin = getattr(self.processor, "packed_pin_memory", lambda: False)()
        if enable is None or not enable(max_len, pin, s.engine_batch_size):
            self._log.warning(
                "engine_packed_mode: listener cannot enable the packed path "
                "(TLS/ws/inproc or extension missing); using the frame loop"
            )
            return False
        # pipelined mode: overlap batch N's async GPU work with batch
        # N+1's recv+decode (submit launches without syncing; collect
        # does the one readback). Falls back to the synchronous call
        # when the component lacks the split API.
        submit = getattr(self.processor, "submit_packed_frames", None)
        collect = getattr(self.processor, "collect_packed_frames", None)
        pipelined = submit is not None and collect is not None
        self._log.info(
            "engine PACKED loop started on %s (max_len=%d pin=%s pipelined=%s)",
            s.engine_addr, max_len, pin, pipelined)
        stats_on = os.environ.get("DMX_ENGINE_STATS") == "1"
        st = {"n": 0, "frames": 0, "recv": 0.0, "proc": 0.0,
              "submit": 0.0, "drain": 0.0, "last": time.perf_counter()}
        prev = None          # in-flight token
        prev_meta = None     # (B, nbytes, t_submit, conn, conn_segs)
        reply_conn = getattr(self._pair_sock, "reply_conn", None)

        def emit(alerts, conn=None, segs=None):
            # segs = [(conn, row_count)] captured AT RECV TIME for this
            # batch (the listener's own state may already describe the
            # NEXT batch when pipelining): merged multi-peer chunks route
            # each alert row to ITS sender.
            for idx, out in alerts:
                if self._out_socks:
                    self._send_to_outputs(out)
                else:
                    target = conn
                    if segs is not None and len(segs) > 1:
                        base = 0
                        for c, rows in segs:
                            if idx < base + rows:
                                target = c
                                break
                            base += rows
                    ok = (reply_conn(target, out) if reply_conn is not None
                          else self._pair_sock.send(out, block=False))
                    if ok:
                        m.data_written_bytes_total.inc(len(out))
                        m.data_written_lines_total.inc(1)
                    else:
                        m.data_dropped_bytes_total.inc(len(out))
                        m.data_dropped_lines_total.inc(1)

        def drain_prev():
            nonlocal prev, prev_meta
            if prev is None:
                return
            pB, pbytes, t_sub, pconn, psegs = prev_meta
            t_d0 = time.perf_counter() if stats_on else 0.0
            try:
                emit(collect(prev), pconn, psegs)
            except Exception as exc:  # noqa: BLE001
                m.processing_errors_total.inc(pB)
                self._log.error("processing error on packed batch of %d: %s",
                                pB, exc)
                prev, prev_meta = None, None
                return
            m.data_processed_bytes_total.inc(pbytes)
            m.data_processed_lines_total.inc(pB)
            m.observe_batch(time.perf_counter() - t_sub, pB)
            if stats_on:
                st["n"] += 1
                st["frames"] += pB
                st["proc"] += time.perf_counter() - t_sub
                st["drain"] += time.perf_counter() - t_d0
            prev, prev_meta = None, None

        while not self._stop_event.is_set():
            try:
                t_r0 = time.perf_counter()
                conn, lines, lens, blob, off, nbytes = (
                    self._pair_sock.recv_packed(
                        s.engine_recv_timeout, s.engine_batch_size,
                        s.engine_batch_linger_ms,
                    )
                )
                if stats_on:
                    st["recv"] += time.perf_counter() - t_r0
            except RecvTimeout:
                drain_prev()  # idle: finish the in-flight batch
                continue
            except SocketClosed:
                break
            except OSError as exc:
                if self._stop_event.is_set():
                    break
                self._log.error("recv error: %s", exc)
                continue
            B = int(lines.shape[0])
            if B == 0:
                drain_prev()
                continue
            m.data_read_bytes_total.inc(nbytes)
            m.data_read_lines_total.inc(B)
            m.engine_batch_size.observe(B)
            segs = getattr(self._pair_sock, "_batch_conn_segs", None)
            t0 = time.perf_counter()
            if pipelined:
                drain_prev()  # collect batch N (GPU overlapped our recv)
                try:
                    prev = submit(lines, lens, blob, off)
                    prev_meta = (B, nbytes, t0, conn, segs)
                    if stats_on:
                        st["submit"] += time.perf_counter() - t0
                except Exception as exc:  # noqa: BLE001
                    m.processing_errors_total.inc(B)
                    self._log.error(
                        "processing error on packed batch of %d: %s", B, exc)
                    prev, prev_meta = None, None
                # sparse traffic: if nothing else is already queued,
                # finish this batch NOW instead of waiting out the recv
                # timeout (keeps single-frame round-trip latency low;
                # under load the queue is non-empty and pipelining holds)
                has_pending = getattr(self._pair_sock, "has_pending", None)
                if has_pending is None or not has_pending():
                    drain_prev()
            else:
                try:
                    alerts = proc(lines, lens, blob, off)
                except Exception as exc:  # noqa: BLE001 - loop must survive
                    m.processing_errors_total.inc(B)
                    self._log.error(
                        "processing error on packed batch of %d: %s", B, exc)
                    continue
                elapsed = time.perf_counter() - t0
                m.data_processed_bytes_total.inc(nbytes)
                m.data_processed_lines_total.inc(B)
                m.observe_batch(elapsed, B)
                emit(alerts, conn, segs)
                if stats_on:
                    st["n"] += 1
                    st["frames"] += B
                    st["proc"] += elapsed
            if stats_on:
                now = time.perf_counter()
                if now - st["last"] > 2.0 and st["n"]:
                    self._log.info(
                        "[packed-stats] %d batches (%.0f fr/b): recv-wait "
                        "%.1fms/b process %.1fms/b "
                        "(submit %.1f drain %.1f)",
                        st["n"], st["frames"] / st["n"],
                        st["recv"] * 1e3 / st["n"], st["proc"] * 1e3 / st["n"],
                        st["submit"] * 1e3 / st["n"], st["drain"] * 1e3 / st["n"],
                    )
                    st.update(n=0, frames=0, recv=0.0, proc=0.0,
                              submit=0.0, drain=0.0, last=now)
        drain_prev()
        self._log.info("engine packed loop exited")
        return True


    # ------------------------------------------------------------------
    # settings-driven distributed placement (dist_mode fanout/stage):
    # the reference wires its parallelism entirely through config
    # (container/config/parser_settings.yaml out_addr); here torchrun +
    # one YAML places the same topology across ranks/GPUs with RCCL
    # collectives (gloo on CPU) instead of socket hops.
    # ------------------------------------------------------------------
    def _run_dist_loop(self) -> None:
        import torch
        import torch.distributed as tdist

        from ..parallel import dist as dmx_dist

        ctx = self._dist_ctx
        s = self.settings
        m = self.metrics
        device = torch.device("cpu")
        if tdist.get_backend() == "nccl":
            device = torch.device("cuda", torch.cuda.current_device())
        mode, rank, world = ctx.mode, ctx.rank, ctx.world
        src = ctx.src if mode == "fanout" else 0
        head = rank == src
        self._log.info(
            "engine DIST loop started: mode=%s rank=%d/%d src=%d head=%s",
            mode, rank, world, src, head)
        if head:
            degraded = self._dist_head_loop(dmx_dist, mode, rank, world,
                                            device)
        else:
            degraded = self._dist_sink_loop(dmx_dist, mode, rank, world,
                                            src, device)
        if degraded and not self._stop_event.is_set():
            # elastic degradation (SURVEY §5.8 hard part): a dead peer
            # broke the collective path — keep serving through the plain
            # socket loop (own engine_addr in, out_addr fan-out, retry-
            # then-drop) instead of going dark. Peers that come back
            # rejoin at the next service restart (communicator reform is
            # parallel/elastic.py's mechanism).
            self._log.warning("dist peer lost: degrading to the socket loop")
            self.metrics.engine_dist_degraded.inc()
            return self._run_socket_loop()
        self._log.info("engine dist loop exited")

    def _dist_head_loop(self, dmx_dist, mode, rank, world, device) -> bool:
        """Returns True when the collective path failed (degrade)."""
        s, m = self.settings, self.metrics
        degraded = False
        while not self._stop_event.is_set():
            try:
                frames = self._pair_sock.recv_many(
                    s.engine_batch_size, s.engine_recv_timeout,
                    s.engine_batch_linger_ms)
            except RecvTimeout:
                frames = []
            except SocketClosed:
                break
            frames = [f for f in frames if f]
            outs: List[bytes] = []
            if frames:
                n_bytes = sum(len(f) for f in frames)
                m.data_read_bytes_total.inc(n_bytes)
                m.data_read_lines_total.inc(len(frames))
                t0 = time.perf_counter()
                try:
                    outs = [o for o in self.processor.process_batch(frames)
                            if o is not None]
                except Exception as exc:  # noqa: BLE001
                    m.processing_errors_total.inc(len(frames))
                    self._log.error("dist processing error: %s", exc)
                    outs = []
                m.data_processed_bytes_total.inc(n_bytes)
                m.data_processed_lines_total.inc(len(frames))
                m.observe_batch(time.perf_counter() - t0, max(len(frames), 1))
            flag = dmx_dist.FRAME_DATA if outs else dmx_dist.FRAME_HEARTBEAT
            try:
                if mode == "fanout":
                    dmx_dist.broadcast_frames_src(outs, rank, device, flag)
                else:
                    dmx_dist.send_frames(outs, rank + 1, device, flag)
            except RuntimeError as exc:
                self._log.error("dist send failed (peer lost?): %s", exc)
                degraded = True
                break
            if outs:
                nb = sum(len(o) for o in outs)
                m.data_written_bytes_total.inc(nb)
                m.data_written_lines_total.inc(len(outs))
        # wake every peer out of its collective before exiting
        try:
            if mode == "fanout":
                dmx_dist.broadcast_frames_src([], rank, device,
                                              dmx_dist.FRAME_SHUTDOWN)
            else:
                dmx_dist.send_frames([], rank + 1, device,
                                     dmx_dist.FRAME_SHUTDOWN)
        except RuntimeError:
            pass
        return degraded

    def _dist_sink_loop(self, dmx_dist, mode, rank, world, src,
                        device) -> bool:
        """Returns True when the collective path failed (degrade)."""
        m = self.metrics
        last = mode == "fanout" or rank == world - 1
        degraded = False
        while not self._stop_event.is_set():
            try:
                if mode == "fanout":
                    frames, flag = dmx_dist.broadcast_frames_sink(src, device)
                else:
                    frames, flag = dmx_dist.recv_frames(rank - 1, device)
            except RuntimeError as exc:
                self._log.error("dist recv failed (peer lost?): %s", exc)
                degraded = True
                break
            if flag == dmx_dist.FRAME_SHUTDOWN:
                if mode == "stage" and rank + 1 < world:
                    try:
                        dmx_dist.send_frames([], rank + 1, device,
                                             dmx_dist.FRAME_SHUTDOWN)
                    except RuntimeError:
                        pass
                break
            outs: List[bytes] = []
            if frames:
                n_bytes = sum(len(f) for f in frames)
                m.data_read_bytes_total.inc(n_bytes)
                m.data_read_lines_total.inc(len(frames))
                t0 = time.perf_counter()
                try:
                    outs = [o for o in self.processor.process_batch(frames)
                            if o is not None]
                except Exception as exc:  # noqa: BLE001
                    m.processing_errors_total.inc(len(frames))
                    self._log.error("dist processing error: %s", exc)
                    outs = []
                m.data_processed_bytes_total.inc(n_bytes)
                m.data_processed_lines_total.inc(len(frames))
                m.observe_batch(time.perf_counter() - t0, max(len(frames), 1))
            if mode == "stage" and not last:
                flag = (dmx_dist.FRAME_DATA if outs
                        else dmx_dist.FRAME_HEARTBEAT)
                try:
                    dmx_dist.send_frames(outs, rank + 1, device, flag)
                except RuntimeError as exc:
                    self._log.error("dist forward failed: %s", exc)
                    degraded = True
                    break
                if outs:
                    m.data_written_bytes_total.inc(sum(len(o) for o in outs))
                    m.data_written_lines_total.inc(len(outs))
            else:
                for out in outs:
                    self._send_to_outputs(out)
        return degraded

    def _run_source_loop(self) -> None:
        """Source mode: the component generates frames (reader services).

        The input socket stays bound (admin/back-channel parity) but the
        data flows component → outputs."""
        s = self.settings
        m = self.metrics
        gen = self.processor.source_batches(s.engine_batch_size, self._stop_event)
        self._log.info("engine source loop started (outputs=%d)", len(self._out_socks))
        for batch in gen:
            if self._stop_event.is_set():
                break
            t0 = time.perf_counter()
            n_bytes = sum(len(f) for f in batch)
            n_lines = sum(_count_lines(f) for f in batch)
            m.data_read_bytes_total.inc(n_bytes)
            m.data_read_lines_total.inc(n_lines)
            m.data_processed_bytes_total.inc(n_bytes)
            m.data_processed_lines_total.inc(n_lines)
            m.engine_batch_size.observe(len(batch))
            for out in batch:
                self._send_to_outputs(out)
            m.observe_batch(time.perf_counter() - t0, n_lines)
        self._log.info("engine source loop exited")

    def _send_to_outputs(self, data: bytes) -> None:
        """Broadcast to all outputs with retry-then-drop per socket
        (reference engine.py:266-302)."""
        m = self.metrics
        for sock in self._out_socks:
            sent = False
            for _attempt in range(self.settings.engine_retry_count):
                try:
                    if sock.send(data, block=False):
                        sent = True
                        break
                except SocketClosed:
                    break
                time.sleep(0.01)
            if sent:
                m.data_written_bytes_total.inc(len(data))
                m.data_written_lines_total.inc(_count_lines(data))
            else:
                m.data_dropped_bytes_total.inc(len(data))
                m.data_dropped_lines_total.inc(_count_lines(data))
                self._log.debug("dropped frame for %s", sock.addr)
