"""The parallel runtime: replica setup, routing ladder, per-step scheduler.

Replaces the reference's monkeypatched ``parallel_forward`` closure stack
(any_device_parallel.py:917-1471). Behavioral parity points:

- routing ladder (any_device_parallel.py:1295-1346):
    batch == 1 and workload_split  -> pipeline (block-sharded) mode
    batch < #devices or split off  -> lead-only
    otherwise                      -> data-parallel split
- split sizes from static weights or the VRAM balancer (:1318-1322),
  size-0 devices skipped (:1324-1337).
- OOM during setup drops the device and renormalizes over survivors; zero
  survivors raises (:1114-1128). Runtime OOM falls back to lead-only
  (:1435-1446). Worker errors carry device attribution (:1424-1427).

MI355X-native differences (SURVEY.md §2 component #5):

- ONE launcher thread enqueues all per-GPU work on persistent per-device HIP
  streams — the reference's ThreadPoolExecutor (one Python thread per device,
  :1414) serializes kernel launches on the GIL; a single thread launching
  into N streams does not.
- Cross-device ordering is by HIP events, not torch.cuda.synchronize: the
  reference fully drains every device's stream before AND after each
  per-step forward (:1385-1391), which forbids any scatter/compute overlap.
- All GPU-GPU movement is direct peer copy over xGMI (stream-ordered
  hipMemcpyPeerAsync underneath), never staged through host memory.
"""
from __future__ import annotations

import contextlib
import logging
from types import MethodType
from typing import Any, Dict, List, Optional

import torch
from torch import nn

from .balance import AdaptiveBalancer, auto_split_batch
from .chain import DeviceChain
from .replicate import replicate_module
from .split import (
    active_split,
    compute_split_sizes,
    concatenate_results,
    get_batch_size,
    move_to_device,
    split_batch,
    split_kwargs,
)

log = logging.getLogger("parallelanything")


def _is_oom(err: BaseException) -> bool:
    if isinstance(err, torch.cuda.OutOfMemoryError):
        return True
    return isinstance(err, RuntimeError) and "out of memory" in str(err).lower()


class WorkerError(RuntimeError):
    """A per-device failure with device attribution
    (reference: any_device_parallel.py:1424-1427)."""

    def __init__(self, device: str, original: BaseException):
        super().__init__(f"worker on {device} failed: {original!r}")
        self.device = device
        self.original = original


class ParallelEngine:
    """Replica set + per-device streams + the per-step scheduler."""

    def __init__(
        self,
        chain: DeviceChain,
        workload_split: bool = True,
        auto_vram_balance: bool = True,
        use_hip_graphs: bool = False,
    ):
        self.chain = chain
        self.workload_split = workload_split
        self.auto_vram_balance = auto_vram_balance
        self.replicas: Dict[str, nn.Module] = {}
        self.streams: Dict[str, Optional[torch.cuda.Stream]] = {}
        self.pipeline = None  # set by pipeline.configure_pipeline
        # shape-keyed hipGraph capture of per-replica forwards (extension;
        # see hipgraphs.py). Off by default: eager == reference behavior.
        self.graphs = None
        if use_hip_graphs:
            from .hipgraphs import GraphRunner

            self.graphs = GraphRunner()
        self._lead_t = torch.device(chain.lead)
        # per-step timing feedback (closes the reference's static-balancing
        # limitation); engaged only when auto_vram_balance is on
        self.balancer: Optional[AdaptiveBalancer] = None
        self._pending_times: list = []

    # ------------------------------------------------------------------
    # Setup: replicate with OOM degradation.
    # ------------------------------------------------------------------
    def setup(self, model: nn.Module, force_copy_lead: bool = False) -> None:
        """Clone ``model`` onto every chain device.

        ``force_copy_lead`` mirrors the reference's clone-even-the-original
        rule when LoRA patches are live (any_device_parallel.py:1073-1081).
        OOM on a device drops it from the chain and renormalizes
        (:1114-1128); any other clone failure rolls back and re-raises
        (:1138-1150).
        """
        chain = self.chain
        idx = 0
        while idx < len(chain.devices):
            dev = chain.devices[idx]
            force = force_copy_lead and idx == 0
            if dev in self.replicas:
                # duplicate device entries (e.g. [cpu, cpu]) share one replica
                idx += 1
                continue
            try:
                replica = replicate_module(model, dev, force_copy=force)
                self.replicas[dev] = replica
                self.streams[dev] = (
                    torch.cuda.Stream(device=dev)
                    if torch.device(dev).type == "cuda"
                    else None
                )
                idx += 1
            except Exception as err:  # noqa: BLE001
                if _is_oom(err):
                    log.warning("OOM replicating to %s; dropping device", dev)
                    if torch.device(dev).type == "cuda":
                        with torch.cuda.device(dev):
                            torch.cuda.empty_cache()
                    if len(chain.devices) <= 1:
                        raise RuntimeError(
                            "no devices survived replication"
                        ) from err
                    chain = chain.drop(idx)
                    self.chain = chain
                else:
                    self.release()
                    raise
        if not self.replicas:
            raise RuntimeError("no devices survived replication")

    @property
    def lead(self) -> str:
        return self.chain.lead

    @property
    def lead_replica(self) -> nn.Module:
        return self.replicas[self.lead]

    # ------------------------------------------------------------------
    # Per-step forward.
    # ------------------------------------------------------------------
    @torch.no_grad()
    def forward(self, x, timesteps, context=None, **kwargs):
        batch = get_batch_size(x)
        devices, weights = list(self.chain.devices), list(self.chain.weights)

        # Routing ladder (reference :1295-1346). Batches too small for DP
        # also take the pipeline when micro-batching is enabled (extension;
        # the reference sends 1 < batch < n_devices to the lead only).
        if (
            self.workload_split
            and self.pipeline is not None
            and (
                batch == 1
                or (batch < len(devices) and self.pipeline.microbatches > 1)
            )
        ):
            return self.pipeline.forward(x, timesteps, context, **kwargs)
        if not self.workload_split or batch < len(devices) or len(devices) == 1:
            return self._lead_only(x, timesteps, context, **kwargs)

        self._harvest_times()
        if self.auto_vram_balance and self.balancer is not None and all(
            self.balancer.throughput.get(d) is not None for d in devices
        ):
            sizes = self.balancer.split(batch)
        elif self.auto_vram_balance:
            sizes = auto_split_batch(batch, devices, weights)
        else:
            sizes = compute_split_sizes(batch, weights)
        act_devices, _act_w, act_sizes = active_split(devices, weights, sizes)
        if len(act_devices) == 1:
            return self._lead_only(x, timesteps, context, **kwargs)

        try:
            return self._data_parallel(
                act_devices, act_sizes, batch, x, timesteps, context, kwargs
            )
        except WorkerError:
            raise
        except Exception as err:  # noqa: BLE001
            if _is_oom(err):
                log.warning("runtime OOM in DP step; falling back to lead-only")
                for dev in act_devices:
                    if torch.device(dev).type == "cuda":
                        with torch.cuda.device(dev):
                            torch.cuda.empty_cache()
                return self._lead_only(x, timesteps, context, **kwargs)
            raise

    def _lead_only(self, x, timesteps, context, **kwargs):
        dev = self.lead
        x = move_to_device(x, dev)
        timesteps = move_to_device(timesteps, dev)
        context = move_to_device(context, dev)
        kwargs = {k: move_to_device(v, dev) for k, v in kwargs.items()}
        return self._run_chunk(dev, x, timesteps, context, kwargs)

    # -- DP path -------------------------------------------------------
    def _data_parallel(self, devices, sizes, batch, x, timesteps, context, kwargs):
        from ..utils.profiling import trace_range

        with trace_range("pa::dp_step"):
            return self._data_parallel_impl(
                devices, sizes, batch, x, timesteps, context, kwargs
            )

    def _harvest_times(self):
        if not self._pending_times:
            return
        still_pending = []
        for dev, size, rec in self._pending_times:
            if isinstance(rec, float):  # cpu worker: wall-clock sample
                self.balancer.record(dev, size, rec)
                continue
            ev0, ev1 = rec
            try:
                if not ev1.query():
                    still_pending.append((dev, size, rec))
                    continue
                self.balancer.record(dev, size, ev0.elapsed_time(ev1) / 1000.0)
            except Exception as err:  # noqa: BLE001
                # drop the sample but say so — this feeds the load balancer
                log.debug("balancer timing sample dropped on %s: %r", dev, err)
        self._pending_times = still_pending

    def _data_parallel_impl(self, devices, sizes, batch, x, timesteps, context, kwargs):
        if self.balancer is None and self.auto_vram_balance:
            # Construct from the FULL chain, not this step's active subset:
            # a size-0 device dropped on the first step would otherwise
            # pair user weights with the wrong devices in weights().
            self.balancer = AdaptiveBalancer(
                list(self.chain.devices), list(self.chain.weights)
            )
        x_chunks = split_batch(x, sizes)
        t_chunks = split_batch(timesteps, sizes)
        c_chunks = split_batch(context, sizes) if context is not None else None
        kw_chunks = split_kwargs(kwargs, sizes, batch)

        lead = self.lead
        lead_is_cuda = self._lead_t.type == "cuda"
        lead_stream = torch.cuda.current_stream(self._lead_t) if lead_is_cuda else None
        ready = None
        if lead_is_cuda:
            ready = torch.cuda.Event()
            ready.record(lead_stream)

        results: List[Any] = [None] * len(devices)
        done_events: List[Optional[torch.cuda.Event]] = [None] * len(devices)
        keep_alive: List[Any] = []  # pin async intermediates until gathered

        # Launch phase: one host thread, one stream per GPU. GPU enqueues
        # go FIRST — a synchronous cpu worker early in a hybrid chain must
        # not delay later stream enqueues (it computes while they run).
        for i in self._launch_order(devices):
            dev = devices[i]
            stream = self.streams.get(dev)
            try:
                if stream is None:  # cpu worker: synchronous
                    import time as _time

                    _t0 = _time.perf_counter()
                    out = self._run_chunk(
                        dev,
                        move_to_device(x_chunks[i], dev),
                        move_to_device(t_chunks[i], dev),
                        move_to_device(c_chunks[i], dev) if c_chunks else None,
                        {k: move_to_device(v, dev)
                         for k, v in kw_chunks[i].items()},
                    )
                    if self.balancer is not None:
                        self._pending_times.append(
                            (dev, sizes[i], _time.perf_counter() - _t0)
                        )
                    results[i] = move_to_device(out, lead)
                    continue
                with torch.cuda.stream(stream), self._lead_stream_ctx(lead_stream):
                    if ready is not None:
                        stream.wait_event(ready)
                    xi = move_to_device(x_chunks[i], dev, non_blocking=True)
                    ti = move_to_device(t_chunks[i], dev, non_blocking=True)
                    ci = (
                        move_to_device(c_chunks[i], dev, non_blocking=True)
                        if c_chunks
                        else None
                    )
                    kwi = {
                        k: move_to_device(v, dev, non_blocking=True)
                        for k, v in kw_chunks[i].items()
                    }
                    if ready is not None:
                        # inputs may have been enqueued on the lead stream;
                        # order the worker stream behind them.
                        ev_in = torch.cuda.Event()
                        ev_in.record(lead_stream)
                        stream.wait_event(ev_in)
                    ev0 = None
                    if self.balancer is not None:
                        ev0 = torch.cuda.Event(enable_timing=True)
                        ev0.record(stream)
                    out = self._run_chunk(dev, xi, ti, ci, kwi)
                    if ev0 is not None:
                        ev1 = torch.cuda.Event(enable_timing=True)
                        ev1.record(stream)
                        self._pending_times.append((dev, sizes[i], (ev0, ev1)))
                    out_lead = move_to_device(out, lead, non_blocking=True)
                    ev = torch.cuda.Event()
                    ev.record(stream)
                    done_events[i] = ev
                    results[i] = out_lead
                    keep_alive += [xi, ti, ci, kwi, out]
            except Exception as err:  # noqa: BLE001
                raise WorkerError(dev, err) from err

        # Gather phase: lead stream waits each worker's event, then cat.
        if lead_is_cuda:
            for ev in done_events:
                if ev is not None:
                    lead_stream.wait_event(ev)
            # order every worker stream after the lead-side copies so the
            # caching allocator cannot reuse worker-side intermediates while
            # a cross-device copy (which may have been enqueued on the lead
            # stream) still reads them
            ev_done = torch.cuda.Event()
            ev_done.record(lead_stream)
            for dev in devices:
                s = self.streams.get(dev)
                if s is not None:
                    s.wait_event(ev_done)
        for i, r in enumerate(results):
            if r is None:
                raise WorkerError(devices[i], RuntimeError("missing result"))
        self._record_gather_streams(results, lead_stream)
        out = concatenate_results(results, dim=0)
        del keep_alive
        return out

    def _launch_order(self, devices) -> List[int]:
        """Indices with GPU (stream-backed) devices first, cpu workers last,
        relative order preserved within each class. The cpu worker runs
        synchronously on the launcher thread; putting it last means every
        GPU stream is already busy while it computes."""
        gpu = [i for i, d in enumerate(devices) if self.streams.get(d) is not None]
        cpu = [i for i, d in enumerate(devices) if self.streams.get(d) is None]
        return gpu + cpu

    @staticmethod
    @contextlib.contextmanager
    def _lead_stream_ctx(lead_stream):
        """Pin the lead device's current stream during worker enqueue so
        cross-device copies land on a stream we order with events, whichever
        side the runtime enqueues them on."""
        if lead_stream is None:
            yield
        else:
            with torch.cuda.stream(lead_stream):
                yield

    @staticmethod
    def _record_gather_streams(results, lead_stream):
        if lead_stream is None:
            return
        for r in results:
            tensors = r if isinstance(r, (list, tuple)) else [r]
            for t in tensors:
                if isinstance(t, torch.Tensor) and t.is_cuda:
                    try:
                        t.record_stream(lead_stream)
                    except Exception:  # noqa: BLE001
                        pass

    def _run_chunk(self, dev, xi, ti, ci, kwi):
        replica = self.replicas[dev]
        # The lead replica may BE the installed model (same-device skip-clone
        # aliasing, reference :594-597): call through _original_forward to
        # avoid re-entering the scheduler (reference :1390).
        fwd = getattr(replica, "_original_forward", None) or replica
        if self.graphs is not None:
            from .pipeline import pipeline_mode_active

            # pipeline mode crosses devices inside the forward: a single
            # per-device graph cannot capture that — stay eager there.
            if not pipeline_mode_active():
                return self.graphs.run(fwd, dev, xi, ti, ci, kwi)
        if ci is not None:
            return fwd(xi, ti, context=ci, **kwi)
        return fwd(xi, ti, **kwi)

    # ------------------------------------------------------------------
    def release(self) -> None:
        """Free replicas and streams (engine half of cleanup; see
        cleanup.cleanup_parallel_model for the model-side restore)."""
        self.replicas.clear()
        self.streams.clear()
        self.pipeline = None
        self.balancer = None
        self._pending_times.clear()
        if self.graphs is not None:
            self.graphs.clear()
        if torch.cuda.is_available():
            torch.cuda.empty_cache()


# ---------------------------------------------------------------------------
# Model install / uninstall (the monkeypatch surface, reference :1450-1459).
# ---------------------------------------------------------------------------

def install_parallel_forward(model: nn.Module, engine: ParallelEngine) -> None:
    """Swap ``model.forward`` for the engine's scheduler.

    State attrs keep the reference's names so downstream tooling that looks
    for them keeps working (any_device_parallel.py:1450-1457).
    """
    if getattr(model, "_true_parallel_active", False):
        uninstall_parallel_forward(model)
    model._original_forward = model.forward
    model._true_parallel_active = True
    model._parallel_engine = engine
    model._parallel_replicas = engine.replicas
    model._parallel_devices = tuple(engine.chain.devices)
    model._parallel_streams = engine.streams
    model._parallel_weights = tuple(engine.chain.weights)
    model._auto_vram_balance = engine.auto_vram_balance

    def _forward(self, x, timesteps, context=None, **kwargs):
        return engine.forward(x, timesteps, context=context, **kwargs)

    model.forward = MethodType(_forward, model)


def uninstall_parallel_forward(model: nn.Module) -> None:
    if hasattr(model, "_original_forward"):
        model.forward = model._original_forward
    for attr in (
        "_original_forward",
        "_true_parallel_active",
        "_parallel_engine",
        "_parallel_replicas",
        "_parallel_devices",
        "_parallel_streams",
        "_parallel_weights",
        "_auto_vram_balance",
    ):
        if hasattr(model, attr):
            try:
                delattr(model, attr)
            except AttributeError:
                pass
