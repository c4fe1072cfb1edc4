"""ShuffleEngine: the per-epoch map -> exchange -> reduce pipeline.

MI355X-native redesign of the reference's shuffle driver + map/reduce Ray
tasks (reference: ray_shuffling_data_loader/shuffle.py:51-219). Instead of a
central driver fanning out object-store tasks, every trainer process runs a
symmetric engine worker:

  map     : read my Parquet shard -> pack rows on GPU (io.read_files_packed);
            the packed source block can stay RESIDENT IN HBM (288 GB/GPU)
            so later epochs shuffle at HBM speed with zero disk/host traffic
            ("in-memory shuffling, loads from disk once" is the reference's
            own contract, shuffle.py:46-48 comment).
  exchange: random destination-trainer assignment per row, GPU radix grouping
            (ops.partition_rows), then ONE RCCL all-to-all over xGMI between
            the per-GPU ranks (parallel.fabric.exchange_rows). This replaces
            the reference's object-store all-to-all (shuffle.py:112-123).
  reduce  : multinomial split of the received rows into this rank's reducer
            partitions + full random permutation, executed as a single fused
            gather kernel per partition (ops.gather_rows / unpack_permute —
            the reference's pd.concat + sample(frac=1) + convert_to_tensor,
            shuffle.py:192-194, torch_dataset.py:204-236).

Statistical equivalence with the reference's two-stage randomness
(per-row uniform reducer id, then per-reducer permutation): partitioning a
uniform random permutation at multinomial(N, 1/R) boundaries yields the same
joint distribution over (partition membership, within-partition order).

The engine runs in a daemon thread per rank; epochs are pipelined under the
consumer's max_concurrent_epochs window (BatchQueue.new_epoch gate), and all
GPU work runs on a dedicated side HIP stream so the trainer's compute stream
never stalls behind shuffle work.
"""

import contextlib
import os
import sys
import threading
import time
from typing import Dict, List, Optional, Sequence, Tuple

_VERBOSE = os.environ.get("RSDL_VERBOSE") == "1"


def _vlog(msg: str) -> None:
    if _VERBOSE:
        print(f"[rsdl-engine] {msg}", file=sys.stderr, flush=True)

import numpy as np
import torch

from ray_shuffling_data_loader_amd.io import (
    fuse_schema,
    infer_schema,
    read_files_packed,
)
from ray_shuffling_data_loader_amd.ops.shuffle_ops import (
    gather_rows,
    partition_rows,
    unpack_permute,
)
from ray_shuffling_data_loader_amd.parallel import fabric
from ray_shuffling_data_loader_amd.utils.rowblock import RowBlock
from ray_shuffling_data_loader_amd.utils.schema import (
    Schema,
    dtype_bytes,
    homogeneous_dtype,
)


class ShuffleEngineFailure:
    """Marker pushed into the batch queues when the engine worker dies, so
    blocked consumers raise instead of hanging (the reference had no analog:
    a dead Ray shuffle task surfaced only at the final ray.get,
    dataset.py:186-188, while iterating trainers hung)."""

    __slots__ = ("error",)

    def __init__(self, error: BaseException):
        self.error = error


class ShuffleEngine:
    def __init__(
        self,
        filenames: Sequence[str],
        consumer,  # BatchConsumer
        num_epochs: int,
        num_reducers: int,
        num_trainers: int,
        rank: int = 0,
        start_epoch: int = 0,
        device: Optional[torch.device] = None,
        seed: Optional[int] = None,
        source_cache: str = "auto",  # "auto" | "device" | "host" | "none"
        reader_threads: int = 8,
        group=None,
        schema: Optional[Schema] = None,
        feature_matrix: Optional[Tuple[str, List[str]]] = None,
        output: str = "auto",  # "auto" | "views" | "columns"
        out_dtypes: Optional[Dict[str, torch.dtype]] = None,
        stats_collector=None,
    ):
        self.filenames = list(filenames)
        self.consumer = consumer
        self.num_epochs = num_epochs
        self.start_epoch = start_epoch
        self.num_reducers = num_reducers
        self.num_trainers = num_trainers
        self.group = group
        self.stats = stats_collector
        self.reader_threads = reader_threads
        self.output = output
        self.out_dtypes = out_dtypes or {}
        self.source_cache = source_cache

        world, dist_rank, initialized = fabric.dist_info()
        # RSDL_FORCE_COLLECTIVE=1 runs the full collective path even at
        # world 1 (self-exchange) — lets a single GPU exercise the exact
        # multi-GPU code path end to end.
        force = os.environ.get("RSDL_FORCE_COLLECTIVE") == "1"
        self.distributed = initialized and (world > 1 or force)
        if self.distributed:
            if num_trainers != world:
                raise ValueError(
                    f"num_trainers={num_trainers} must equal "
                    f"torch.distributed world size {world}"
                )
            if rank != dist_rank:
                raise ValueError(
                    f"rank={rank} disagrees with torch.distributed rank "
                    f"{dist_rank}"
                )
            if self.group is None:
                # Dedicated communicator: the engine's collectives run on a
                # background thread and must not interleave with trainer
                # collectives (DDP allreduce) on the default group.
                # (__init__ runs on the main thread on every rank.)
                self.group = fabric.get_shuffle_group()
        self.rank = rank

        if device is None:
            device = torch.device(
                "cuda" if torch.cuda.is_available() else "cpu"
            )
        self.device = torch.device(device)

        # File shard: symmetric striped split in distributed mode; the whole
        # set in local mode (reference: one map task per file, any node —
        # shuffle.py:111-114).
        if self.distributed:
            self.my_files = self.filenames[self.rank :: num_trainers]
            self.owned_trainers = [self.rank]
        else:
            self.my_files = self.filenames
            self.owned_trainers = list(range(num_trainers))

        # Reducer -> trainer ownership: np.array_split parity
        # (reference shuffle.py:125-126).
        splits = np.array_split(np.arange(num_reducers), num_trainers)
        self.reducers_per_trainer = [len(s) for s in splits]
        if min(self.reducers_per_trainer) == 0:
            raise ValueError(
                f"num_reducers={num_reducers} < num_trainers={num_trainers}"
            )

        # Destination-trainer sampling weights proportional to owned reducers
        # (uniform over reducers overall).
        probs = np.array(self.reducers_per_trainer, dtype=np.float64)
        probs /= probs.sum()
        self._dest_uniform = len(set(self.reducers_per_trainer)) == 1
        self._dest_cum = torch.from_numpy(np.cumsum(probs)).to(
            self.device, torch.float32
        )

        if schema is None:
            if not self.filenames:
                raise ValueError("need filenames or an explicit schema")
            schema = infer_schema(self.filenames[0])
        self.base_schema = schema
        self.out_schema = fuse_schema(schema, feature_matrix)

        if seed is None:
            seed = int(time.time_ns() % (2**31))
        self.seed = seed
        if self.device.type == "cuda":
            self._gen = torch.Generator(device=self.device)
        else:
            self._gen = torch.Generator()

        self._cached_source: Optional[torch.Tensor] = None
        self._stream = (
            torch.cuda.Stream(device=self.device)
            if self.device.type == "cuda"
            else None
        )
        self._thread: Optional[threading.Thread] = None
        self._error: Optional[BaseException] = None
        self.duration: Optional[float] = None

    # ----- source ------------------------------------------------------------

    def _get_source(self, epoch: int) -> torch.Tensor:
        if self._cached_source is not None:
            src = self._cached_source
            if src.device != self.device:
                src = src.to(self.device, non_blocking=True)
            return src
        if self.stats:
            # One map "task" per file (reference shuffle.py:111-114).
            for _ in self.my_files:
                self.stats.map_start(epoch)
        t0 = time.perf_counter()
        src = read_files_packed(
            self.my_files, self.base_schema, self.device, self.reader_threads
        )
        read_dur = time.perf_counter() - t0
        if self.stats:
            per_file = read_dur / max(1, len(self.my_files))
            for _ in self.my_files:
                self.stats.map_done(epoch, per_file, per_file)
        if self.source_cache in ("auto", "device"):
            self._cached_source = src
        elif self.source_cache == "host":
            if src.device.type == "cuda":
                # pinned host cache: the per-epoch re-upload then runs as
                # an async DMA instead of a pageable copy
                host = torch.empty(
                    src.shape, dtype=src.dtype, pin_memory=True
                )
                host.copy_(src)
                self._cached_source = host
            else:
                self._cached_source = src
        return src

    # ----- epoch pipeline -----------------------------------------------------

    def _sample_dest(self, n: int) -> torch.Tensor:
        if self._dest_uniform:
            return torch.randint(
                self.num_trainers,
                (n,),
                device=self.device,
                generator=self._gen,
                dtype=torch.long,
            )
        u = torch.rand(n, device=self.device, generator=self._gen)
        return torch.searchsorted(self._dest_cum, u, right=True).clamp_(
            max=self.num_trainers - 1
        )

    def _make_partition(
        self, rows: torch.Tensor, idx: torch.Tensor
    ) -> RowBlock:
        """One reducer partition: fused gather-permute (+ cast/pack) of the
        selected rows into a RowBlock."""
        schema = self.out_schema
        hom = homogeneous_dtype(schema)
        # Output dtype casts (e.g. fp32 -> bf16 features) happen inside the
        # fused unpack kernel, which requires the columns path.
        use_views = not self.out_dtypes and (
            self.output == "views"
            or (self.output == "auto" and hom is not None)
        )
        if use_views and hom is not None:
            esz = dtype_bytes(hom)
            gathered = gather_rows(rows, idx)
            typed = gathered.view(hom)  # [n, row_stride/esz]
            cols: Dict[str, torch.Tensor] = {}
            for spec in schema.columns:
                o = schema.offsets[spec.name] // esz
                if spec.numel == 1:
                    cols[spec.name] = typed[:, o]
                else:
                    cols[spec.name] = typed[:, o : o + spec.numel]
            return RowBlock(cols)
        cols = unpack_permute(
            rows, schema, perm=idx.to(torch.long), out_dtypes=self.out_dtypes
        )
        return RowBlock(cols)

    def _epoch_seed(self, epoch: int) -> int:
        # Deterministic per-(seed, rank, epoch) stream: epoch k's shuffle is
        # independent of which epochs ran before it, so a resumed run
        # (start_epoch=k, same seed) reproduces exactly the shuffles the
        # original run would have produced. (The reference has no resume
        # support at all — SURVEY §5 checkpoint/resume: none.)
        return (self.seed * 1000003 + self.rank) ^ (epoch * 0x9E3779B1)

    def _shuffle_epoch(self, epoch: int) -> None:
        self._gen.manual_seed(self._epoch_seed(epoch) % (2**63))
        if self.stats:
            self.stats.epoch_start(epoch)
        t0 = time.perf_counter()
        src = self._get_source(epoch)
        n = src.shape[0]

        # --- map side: destination assignment + grouping + exchange --------
        if self.distributed:
            dest = self._sample_dest(n)
            grouped, send_counts = partition_rows(
                src, dest, self.num_trainers
            )
            del dest
            _vlog(
                f"rank {self.rank} epoch {epoch}: exchanging "
                f"{int(grouped.shape[0])} rows"
            )
            recv, _ = fabric.exchange_rows(
                grouped, send_counts, self.group
            )
            # Free the send staging (one full shard) before the reduce-side
            # gathers allocate the partitions.
            del grouped
            _vlog(
                f"rank {self.rank} epoch {epoch}: received "
                f"{int(recv.shape[0])} rows"
            )
            rows_per_trainer = {self.rank: recv}
        elif self.num_trainers == 1:
            rows_per_trainer = {0: src}
        else:
            dest = self._sample_dest(n)
            grouped, counts = partition_rows(src, dest, self.num_trainers)
            offs = torch.zeros(
                self.num_trainers + 1, dtype=torch.long
            )
            torch.cumsum(counts.cpu(), 0, out=offs[1:])
            rows_per_trainer = {
                t: grouped[offs[t] : offs[t + 1]]
                for t in range(self.num_trainers)
            }

        # --- reduce side: per owned trainer, multinomial reducer split +
        #     fused permute-gather ------------------------------------------
        for t in self.owned_trainers:
            rows = rows_per_trainer[t]
            m = rows.shape[0]
            r_t = self.reducers_per_trainer[t]
            # multinomial sizes == counts of iid uniform reducer assignment
            assign = torch.randint(
                r_t,
                (m,),
                device=self.device,
                generator=self._gen,
            )
            sizes = torch.bincount(assign, minlength=r_t).cpu()
            perm = torch.randperm(
                m, device=self.device, generator=self._gen
            )
            offs = [0]
            for s in sizes.tolist():
                offs.append(offs[-1] + s)
            parts = []
            for k in range(r_t):
                if self.stats:
                    self.stats.reduce_start(epoch)
                rt0 = time.perf_counter()
                idx = perm[offs[k] : offs[k + 1]]
                parts.append(self._make_partition(rows, idx))
                if self.stats:
                    self.stats.reduce_done(
                        epoch, time.perf_counter() - rt0
                    )
            if self._stream is not None:
                # Partitions are produced on the side stream; make them safe
                # for the consumer's stream before they enter the queue.
                torch.cuda.current_stream(self.device).synchronize()
            self.consumer.consume(t, epoch, parts)
            self.consumer.producer_done(t, epoch)
            _vlog(
                f"rank {self.rank} epoch {epoch}: queued "
                f"{len(parts)} partitions for trainer {t} "
                f"({time.perf_counter() - t0:.3f}s since epoch start)"
            )
        if self.stats:
            self.stats.epoch_done(epoch, time.perf_counter() - t0)

    # ----- driver loop --------------------------------------------------------

    def run(self) -> float:
        """The shuffle driver loop (reference shuffle.py:51-86): per epoch,
        gate on the consumer's epoch window, then shuffle."""
        start = time.perf_counter()
        ctx = (
            torch.cuda.stream(self._stream)
            if self._stream is not None
            else contextlib.nullcontext()
        )
        with ctx:
            for epoch in range(self.start_epoch, self.num_epochs):
                t_gate = time.perf_counter()
                self.consumer.wait_until_ready(epoch)
                if self.stats:
                    # Epoch-window throttle time (reference shuffle.py:74 +
                    # stats.py:166-168 throttle_done).
                    self.stats.epoch_throttle_done(
                        epoch, time.perf_counter() - t_gate
                    )
                self._shuffle_epoch(epoch)
        self.consumer.wait_until_all_epochs_done()
        self.duration = time.perf_counter() - start
        if self.stats:
            self.stats.trial_done(self.duration)
        return self.duration

    def start(self) -> None:
        """Run the driver loop on a background thread."""

        def _target():
            try:
                self.run()
            except BaseException as e:  # surfaced via join() AND the queue
                self._error = e
                failure = ShuffleEngineFailure(e)
                # Per-(epoch, trainer) scope: delivering into an epoch that
                # was already consumed/evicted (or a shut-down queue) may
                # raise, and that must not stop the LATER epochs — the ones
                # consumers are actually blocked on — from getting their
                # failure marker.
                for epoch in range(self.start_epoch, self.num_epochs):
                    for t in self.owned_trainers:
                        try:
                            self.consumer.consume(t, epoch, [failure])
                            self.consumer.producer_done(t, epoch)
                        except BaseException:
                            pass

        self._thread = threading.Thread(
            target=_target, name=f"rsdl-shuffle-r{self.rank}", daemon=True
        )
        self._thread.start()

    def join(self, timeout: Optional[float] = None) -> None:
        if self._thread is not None:
            self._thread.join(timeout)
            if self._error is not None:
                raise RuntimeError(
                    "shuffle engine worker failed"
                ) from self._error

