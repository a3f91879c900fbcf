"""DistriConfig — topology + algorithm knobs for displaced patch parallelism.

Semantics parity with the reference config object
(/root/reference/distrifuser/utils.py:23-109) but re-designed for MI355X:

* one process per GPU over RCCL (``torch.distributed`` backend "nccl" IS
  RCCL on ROCm); falls back to "gloo" on CPU so the whole control plane is
  testable without a GPU;
* the CFG batch is split across the two rank halves and each half splits
  the latent into row patches over xGMI peers (7 point-to-point links per
  GPU — the patch group is the unit all stale-activation all-gathers run
  on);
* knob names match the reference so scripts/configs carry over 1:1.
"""

from __future__ import annotations

import os

import torch
import torch.distributed as dist

VALID_MODES = (
    "separate_gn",
    "stale_gn",
    "corrected_async_gn",
    "sync_gn",
    "full_sync",
    "no_sync",
)
VALID_PARALLELISM = ("patch", "tensor", "naive_patch")
VALID_SPLIT_SCHEMES = ("row", "col", "alternate")


def is_power_of_2(n: int) -> bool:
    return n >= 1 and (n & (n - 1)) == 0


def check_env() -> None:
    """Refuse stacks that cannot run collectives inside HIP graphs.

    The reference required CUDA>=11.3/torch>=2.2 for NCCL-inside-CUDA-graph
    (reference utils.py:6-16). Our floor: torch >= 2.2 with ROCm (or CPU for
    tests).
    """
    major, minor = (int(x) for x in torch.__version__.split(".")[:2])
    if (major, minor) < (2, 2):
        raise RuntimeError(
            f"torch >= 2.2 required for collectives captured in HIP graphs, got {torch.__version__}"
        )
    if torch.cuda.is_available() and torch.version.hip is None:
        raise RuntimeError("distrifuser_amd targets ROCm (gfx950); CUDA builds are unsupported")


class DistriConfig:
    def __init__(
        self,
        height: int = 1024,
        width: int = 1024,
        do_classifier_free_guidance: bool = True,
        split_batch: bool = True,
        warmup_steps: int = 4,
        comm_checkpoint: int = 60,
        mode: str = "corrected_async_gn",
        use_cuda_graph: bool = True,
        parallelism: str = "patch",
        split_scheme: str = "row",
        verbose: bool = False,
        backend: str | None = None,
        device: str | torch.device | None = None,
    ):
        assert mode in VALID_MODES, f"mode must be one of {VALID_MODES}, got {mode!r}"
        assert parallelism in VALID_PARALLELISM, (
            f"parallelism must be one of {VALID_PARALLELISM}, got {parallelism!r}"
        )
        assert split_scheme in VALID_SPLIT_SCHEMES, (
            f"split_scheme must be one of {VALID_SPLIT_SCHEMES}, got {split_scheme!r}"
        )
        check_env()

        self.height = height
        self.width = width
        self.do_classifier_free_guidance = do_classifier_free_guidance
        self.split_batch = split_batch
        self.warmup_steps = warmup_steps
        self.comm_checkpoint = comm_checkpoint
        self.mode = mode
        self.use_cuda_graph = use_cuda_graph
        self.parallelism = parallelism
        self.split_scheme = split_scheme
        self.verbose = verbose

        rank = 0
        world_size = 1
        if dist.is_initialized():
            rank = dist.get_rank()
            world_size = dist.get_world_size()
        else:
            env_world = int(os.environ.get("WORLD_SIZE", "1"))
            if env_world > 1:
                if backend is None:
                    backend = "nccl" if torch.cuda.is_available() else "gloo"
                try:
                    dist.init_process_group(backend=backend)
                    rank = dist.get_rank()
                    world_size = dist.get_world_size()
                except Exception as exc:  # pragma: no cover - rendezvous failure path
                    # Parity with the reference's single-GPU fallback
                    # (reference utils.py:44-47).
                    print(f"[distrifuser_amd] init_process_group failed ({exc}); running single-process")
                    rank, world_size = 0, 1

        assert is_power_of_2(world_size), f"world size must be a power of 2, got {world_size}"

        self.rank = rank
        self.world_size = world_size
        self.local_rank = int(os.environ.get("LOCAL_RANK", rank % max(torch.cuda.device_count(), 1)))

        # CFG batch split doubles total parallelism: half the ranks run the
        # unconditional branch, half the conditional branch; within each half
        # the latent is split into n_device_per_batch row patches.
        if do_classifier_free_guidance and split_batch and world_size >= 2:
            self.n_device_per_batch = world_size // 2
        else:
            self.n_device_per_batch = world_size
            self.split_batch = False

        if device is not None:
            self.device = torch.device(device)
        elif torch.cuda.is_available():
            self.device = torch.device(f"cuda:{self.local_rank}")
        else:
            self.device = torch.device("cpu")
        if self.device.type == "cuda":
            torch.cuda.set_device(self.device)

        # Process-group families (every rank must create every group —
        # dist.new_group is collective):
        #   batch groups: the two CFG halves; the patch peers a rank
        #     all-gathers stale activations with.
        #   split groups: the CFG pair {i, i + ws/2} holding the two guidance
        #     branches of the same patch.
        self._batch_group = None
        self._split_group = None
        if world_size > 1 and dist.is_initialized():
            if self.split_batch:
                half = world_size // 2
                for start in (0, half):
                    ranks = list(range(start, start + half))
                    group = dist.new_group(ranks)
                    if rank in ranks:
                        self._batch_group = group
                for i in range(half):
                    ranks = [i, i + half]
                    group = dist.new_group(ranks)
                    if rank in ranks:
                        self._split_group = group
            else:
                self._batch_group = dist.group.WORLD

    # -- topology ---------------------------------------------------------

    @property
    def batch_group(self):
        """Group of patch peers inside this rank's CFG branch."""
        return self._batch_group

    @property
    def split_group(self):
        """Group pairing the two CFG branches of this rank's patch."""
        return self._split_group

    def batch_idx(self, rank: int | None = None) -> int:
        """Which CFG batch element this rank computes (0 = unconditional)."""
        if rank is None:
            rank = self.rank
        if not self.split_batch:
            return 0
        # Parity with reference semantics (utils.py:98-104): the first half
        # of the ranks computes CFG element 0 (uncond), the second half
        # element 1 (cond).
        return 1 - int(rank < self.world_size // 2)

    def split_idx(self, rank: int | None = None) -> int:
        """This rank's patch index within its CFG branch."""
        if rank is None:
            rank = self.rank
        return rank % self.n_device_per_batch

    # -- conveniences ------------------------------------------------------

    @property
    def use_patch_parallelism(self) -> bool:
        return self.parallelism == "patch" and self.n_device_per_batch > 1

    def __repr__(self) -> str:  # pragma: no cover - debug aid
        return (
            f"DistriConfig(rank={self.rank}/{self.world_size}, device={self.device}, "
            f"parallelism={self.parallelism!r}, mode={self.mode!r}, "
            f"n_device_per_batch={self.n_device_per_batch}, split_batch={self.split_batch}, "
            f"hw={self.height}x{self.width})"
        )
