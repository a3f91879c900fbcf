"""Async stale-activation exchange engine for displaced patch parallelism.

The displaced-patch algorithm (reference PatchParallelismCommManager,
/root/reference/distrifuser/utils.py:112-199) needs every patch peer to see
every other peer's *previous-step* activations (self-attn KV, conv halos,
GroupNorm statistics). The design here keeps the reference's key property —
ONE flat pre-registered buffer per peer so dozens of per-layer tensors ride a
single batched async all-gather — but is laid out for MI355X:

* the flat buffer is a single ``[n_peers, total_numel]`` device tensor so an
  all-gather of any contiguous registration range is a strided write into
  rows, never a torch.cat;
* gathers are issued ``async_op=True`` on RCCL; over xGMI (7 point-to-point
  links per GPU) a batched gather of tens of MB comfortably overlaps with the
  U-Net compute that produced it;
* consumers wait lazily, right before the *next* use of the same tensor one
  denoising step later (1-step staleness).
"""

from __future__ import annotations

import os

import torch
import torch.distributed as dist

from .config import DistriConfig

# Race-debug mode: make every stale-activation gather synchronous
# so any handle-lifecycle bug shows up as a numerics diff, not a race.
DEBUG_SYNC = os.environ.get("DFA_DEBUG_SYNC", "0") == "1"


class PatchParallelismCommManager:
    def __init__(self, distri_config: DistriConfig):
        self.distri_config = distri_config

        self.torch_dtype: torch.dtype | None = None
        self.numels = 0
        self.numels_per_layer_type: dict[str, int] = {}

        self.starts: list[int] = []
        self.ends: list[int] = []
        self.shapes: list[torch.Size] = []

        self.buffer: torch.Tensor | None = None  # [n_peers, total_numel]
        self.handles: list = []

        # batching state
        self.idx_queue: list[int] = []

        # observability: per-collective accounting (bytes pushed per rank);
        # DFA_COMM_TIMING=1 adds per-gather device timing via event pairs
        # (skipped inside hipGraph capture, where events are not allowed)
        self.stats = {"gathers": 0, "bytes": 0, "gather_ms": []}
        self._timing = os.environ.get("DFA_COMM_TIMING", "0") == "1"
        # DFA_COMM_SIDE_STREAM=1 issues the batched gather from a dedicated
        # HIP stream with event handoff (SURVEY §2.4b xGMI mapping). RCCL
        # already runs collectives on its own internal stream, so this
        # mainly decouples the gather ENQUEUE from the compute stream; kept
        # opt-in until a multi-GPU measurement shows a win.
        self._use_side = os.environ.get("DFA_COMM_SIDE_STREAM", "0") == "1"
        self._side_stream: torch.cuda.Stream | None = None
        self._timing_events: list = []

    # -- registration pass -------------------------------------------------

    def register_tensor(
        self,
        shape: tuple[int, ...] | torch.Size,
        torch_dtype: torch.dtype,
        layer_type: str | None = None,
    ) -> int:
        """Reserve a slot in the flat buffer; returns the tensor's idx.

        Called once per wrapped layer during the registration pass, in module
        execution order, so consecutive idxs are contiguous in memory and a
        step's worth of enqueues coalesces into one collective.
        """
        if self.torch_dtype is None:
            self.torch_dtype = torch_dtype
        else:
            assert self.torch_dtype == torch_dtype, (
                f"all registered tensors must share a dtype; "
                f"got {torch_dtype} after {self.torch_dtype}"
            )
        idx = len(self.starts)
        numel = 1
        for d in shape:
            numel *= int(d)
        # 16B-align every slot start so the gfx950 kernels can issue uint4
        # loads straight out of the flat buffer (8 elements at 2B dtypes).
        self.numels = (self.numels + 7) // 8 * 8
        self.starts.append(self.numels)
        self.ends.append(self.numels + numel)
        self.shapes.append(torch.Size(shape))
        self.numels += numel
        if layer_type is not None:
            self.numels_per_layer_type[layer_type] = (
                self.numels_per_layer_type.get(layer_type, 0) + numel
            )
        return idx

    def create_buffer(self) -> None:
        n = self.distri_config.n_device_per_batch
        if self.numels == 0:
            self.handles = []
            return
        # row stride (= per-peer chunk stride seen by the attention kernel)
        # must stay 16B-aligned too
        self.numels = (self.numels + 7) // 8 * 8
        self.buffer = torch.empty(
            (n, self.numels), dtype=self.torch_dtype, device=self.distri_config.device
        )
        self.handles = [None for _ in self.starts]
        if self.distri_config.rank == 0 and self.distri_config.verbose:
            elem = self.buffer.element_size()
            print(
                f"[comm] flat stale-activation buffer: {n} peers x "
                f"{self.numels * elem / 1024 ** 2:.1f} MiB "
                f"({len(self.starts)} tensors)"
            )
            for lt, numel in self.numels_per_layer_type.items():
                print(f"[comm]   {lt}: {numel * elem / 1024 ** 2:.1f} MiB")

    def get_buffer_list(self, idx: int) -> list[torch.Tensor] | None:
        """Per-peer views of slot ``idx``, shaped like the registered tensor."""
        if self.buffer is None:
            return None
        s, e, shape = self.starts[idx], self.ends[idx], self.shapes[idx]
        return [self.buffer[p, s:e].view(shape) for p in range(self.buffer.shape[0])]

    # -- steady-state ------------------------------------------------------

    def enqueue(self, idx: int, tensor: torch.Tensor | None = None) -> None:
        """Publish this layer's fresh activation and batch it for gathering.

        ``tensor=None`` means the producer already staged the fresh data into
        this rank's buffer slot (e.g. self-attention writes its fresh KV
        slice directly into the slot the kernel reads from). Flushes the
        pending batch when the idx wraps (a new denoising step began) or when
        ``comm_checkpoint`` tensors are queued.
        """
        if idx == 0 or (len(self.idx_queue) > 0 and idx <= self.idx_queue[-1]):
            self.communicate()
        assert len(self.idx_queue) == 0 or idx == self.idx_queue[-1] + 1, (
            "enqueue order must match registration order"
        )
        own = self.distri_config.split_idx()
        if tensor is not None and tensor.data_ptr() != self.buffer[own, self.starts[idx]].data_ptr():
            self.buffer[own, self.starts[idx] : self.ends[idx]].copy_(tensor.reshape(-1))
        self.idx_queue.append(idx)
        if len(self.idx_queue) >= self.distri_config.comm_checkpoint:
            self.communicate()

    def communicate(self) -> None:
        """Issue ONE async all-gather covering the whole queued idx range."""
        if not self.idx_queue:
            return
        start = self.starts[self.idx_queue[0]]
        end = self.ends[self.idx_queue[-1]]
        group = self.distri_config.batch_group
        own = self.distri_config.split_idx()
        tensor_list = [self.buffer[p, start:end] for p in range(self.buffer.shape[0])]
        on_gpu = self.buffer.is_cuda
        capturing = on_gpu and torch.cuda.is_current_stream_capturing()
        time_this = self._timing and on_gpu and not capturing
        if time_this:
            ev0 = torch.cuda.Event(enable_timing=True)
            ev1 = torch.cuda.Event(enable_timing=True)

        def _issue():
            if time_this:
                ev0.record()
            h = dist.all_gather(
                tensor_list, self.buffer[own, start:end], group=group,
                async_op=not DEBUG_SYNC,
            )
            if time_this:
                ev1.record()
                self._timing_events.append((ev0, ev1))
            return h

        if self._use_side and on_gpu and not capturing:
            if self._side_stream is None:
                self._side_stream = torch.cuda.Stream()
            ready = torch.cuda.Event()
            ready.record()  # buffer slot writes on the compute stream
            with torch.cuda.stream(self._side_stream):
                self._side_stream.wait_event(ready)
                handle = _issue()
        else:
            handle = _issue()
        self.stats["gathers"] += 1
        self.stats["bytes"] += (end - start) * self.buffer.element_size()
        for idx in self.idx_queue:
            self.handles[idx] = handle if not DEBUG_SYNC else None
        self.idx_queue = []

    def wait(self, idx: int) -> None:
        """Block until slot ``idx``'s last gather has landed (lazy consume)."""
        h = self.handles[idx]
        if h is not None:
            h.wait()
            # One handle covers a contiguous batch; clear every slot sharing it.
            for i, other in enumerate(self.handles):
                if other is h:
                    self.handles[i] = None

    def clear(self) -> None:
        """Flush the queue and drain every outstanding handle."""
        if self.idx_queue:
            self.communicate()
        for i, h in enumerate(self.handles):
            if h is not None:
                h.wait()
                self.handles[i] = None
        self.harvest_timing()

    def harvest_timing(self) -> None:
        """Fold completed per-gather event pairs into stats["gather_ms"]."""
        if not self._timing_events:
            return
        torch.cuda.synchronize()
        for ev0, ev1 in self._timing_events:
            self.stats["gather_ms"].append(ev0.elapsed_time(ev1))
        self._timing_events = []
