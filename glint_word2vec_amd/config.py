"""Configuration surface for the MI355X-native Word2Vec trainer.

Mirrors the full user-visible knob set of the reference
(/root/reference/src/main/scala/org/apache/spark/mllib/feature/ServerSideGlintWord2Vec.scala:67-85,246-251
and the ml.feature Params at ml:40-222), with the parameter-server/HOCON tier
replaced by a device/communication section appropriate for a single 8xMI355X
node.  Defaults are identical to the reference where the knob carries over.
"""
from __future__ import annotations

import dataclasses
import json
from dataclasses import dataclass
from typing import Optional

# Constants carried over from the reference (mllib:246-251).
MAX_EXP = 6.0          # sigmoid clip range [-6, 6] (mllib:248, 281-302)
EXP_TABLE_SIZE = 1000  # only used by the optional LUT-parity sigmoid mode

# Row strides the HIP kernels support (stride = 64 * NC; csrc/hip
# FOR_EACH_NC).  round_stride_py mirrors _hip_native.round_stride so CPU-only
# code (bench policy, memory estimates) agrees with the kernels exactly.
KERNEL_NCS = (1, 2, 3, 4, 5, 6, 8, 10, 12, 16, 20, 24, 32)


def round_stride_py(dim: int) -> int:
    nc_min = (dim + 63) // 64
    for nc in KERNEL_NCS:
        if nc >= nc_min:
            return 64 * nc
    raise ValueError(f"dim {dim} too large (max {64 * KERNEL_NCS[-1]})")


# dp-vs-dim auto-engine threshold: the dp engine allreduces full-table
# deltas every sync_every steps; at bf16 wire dtype a sync moves
# ~2*vocab*stride*2 bytes over ring xGMI (~153 GB/s per link), so tables
# beyond a few GB make the merge cost exceed the step time and the
# dimension-sharded engine (traffic ~4 bytes/pair, dimension-independent)
# wins even though its per-GPU kernel rate is lower.  4 GiB of padded
# bf16 tables ~= vocab 3M at dim 300: sync ~18 ms vs ~12 ms step at
# sync_every=4 — the crossover region.  (288 GB HBM is NOT the binding
# constraint for dp; xGMI merge bandwidth is.)
DP_MAX_TABLE_BYTES = 4 << 30


def choose_engine(vocab_size: int, dim: int, dtype_bytes: int,
                  world: int) -> str:
    """Auto engine policy, shared by engine.py and bench.py (both use the
    padded kernel stride so the estimate matches the real allocation).

    world 1 -> fused.  Small tables -> dp (replicated + overlapped
    delta-allreduce; per-GPU fused rate).  Beyond the dp merge budget ->
    ROW-sharded (the north-star PS shape): data-parallel, per-rank pull
    cycle measured at 63M words/s at vocab 10M / 33M at 80M (world-1
    proxy, benchmarks/results.md) -> ~8x that per node, vs the
    dim-sharded engine whose ranks all walk the SAME batch (~230M/node
    ceiling at the 8-GPU slice shape).  dim remains selectable: its
    traffic is dimension-independent (~4 B/pair), the right tool if
    xGMI, not HBM, turns out to bind."""
    if world == 1:
        return "fused"
    table_bytes = 2 * vocab_size * round_stride_py(dim) * dtype_bytes
    return "dp" if table_bytes <= DP_MAX_TABLE_BYTES else "row"


@dataclass
class Word2VecConfig:
    """All user knobs.  Names are snake_case; the estimator exposes the
    reference's fluent ``setX`` aliases on top of this.

    Reference defaults: vectorSize=100, stepSize=0.01875, numPartitions=1,
    numIterations=1, minCount=5, maxSentenceLength=1000, window=5,
    batchSize=50, n=5, subsampleRatio=1e-6, unigramTableSize=1e8
    (mllib:67-85).
    """

    # --- model shape -------------------------------------------------------
    vector_size: int = 100           # embedding dimension (mllib:67)
    # --- optimisation ------------------------------------------------------
    learning_rate: float = 0.01875   # stepSize (mllib:69)
    num_iterations: int = 1          # epochs (mllib:73)
    # batchSize carried for API parity; the reference's batchSize*n*window
    # <= 10000 constraint was an Akka payload limit (mllib:83-85) that has
    # no xGMI analog — step granularity here is words_per_step/chunk_words.
    batch_size: int = 50             # positions per mini-batch (mllib:79)
    window: int = 5                  # max context window (mllib:77)
    n: int = 5                       # negatives per positive pair (mllib:81)
    subsample_ratio: float = 1e-6    # frequency subsampling ratio (mllib:83)
    seed: Optional[int] = None       # RNG seed (mllib:71); None = random
    # --- vocabulary --------------------------------------------------------
    min_count: int = 5               # (mllib:75)
    max_sentence_length: int = 1000  # sentence chunk length (mllib:88-97)
    unigram_table_size: int = 100_000_000  # negative-sampling table (mllib:85)
    unigram_power: float = 0.75      # classic word2vec table exponent
    # --- parallelism (replaces numPartitions / numParameterServers) --------
    # numPartitions (mllib:120-127) = concurrent async workers.  On the CPU
    # trainer this is the hogwild thread count.  On GPU it is accepted but
    # INERT by design: the reference used it to create concurrency, and the
    # fused kernel already runs ~8k concurrent hogwild waves per launch —
    # the GPU analog of numPartitions is the wave count, not a user knob.
    num_partitions: int = 1
    num_shards: Optional[int] = None  # row shards; None = world size
    # --- device ------------------------------------------------------------
    # table storage dtype. "auto" = bfloat16 on GPU, float32 on CPU: the
    # planted-synonym probe measured bf16 as BOTH faster and higher-quality
    # than fp32 on GPU (implicit regularization — benchmarks/results.md).
    dtype: str = "auto"              # "auto" | "float32" | "bfloat16"
    device: str = "auto"             # "auto" | "cpu" | "cuda"
    words_per_step: int = 1 << 20    # tokens fed to the GPU per training step
    # Row-update mode for the massively concurrent GPU kernels:
    #   "hogwild" — plain read-modify-write: the reference's races-embraced
    #     fire-and-forget semantics (mllib:425); fastest, but lost updates
    #     cap hot-word quality (planted-NN acc plateaus ~0.61,
    #     benchmarks/results.md).
    #   "atomic"  — atomics on every row (fp32 atomicAdd / gfx950 packed
    #     bf16): no lost updates, best quality (acc ~0.81), ~2.5x slower.
    #     On degenerate corpora (hundreds of words, no subsampling) the
    #     summed stale gradients on ultra-hot rows can run away; fine on
    #     real vocabularies.
    #   "hybrid" (default) — atomics only on rows < hybrid_hot_rows (the
    #     vocabulary is sorted by count, so these ARE the contended Zipf
    #     head where hogwild loses updates); the cold tail — rarely
    #     contended, hogwild == atomic there in practice — takes the
    #     cheap path.  Measured on the planted-synonym probe: quality of
    #     full atomics at most of hogwild's speed (benchmarks/results.md).
    # The benchmark and fit() share this default (same semantics measured
    # as shipped).  See DESIGN.md.
    update_mode: str = "hybrid"      # "hogwild" | "atomic" | "hybrid"
    hybrid_hot_rows: int = 32768     # hybrid: atomics for rows < this
    # hybrid: rows < this stay hogwild even inside the atomic head — the
    # ultra-hot top rows take a double-digit share of all negative-table
    # draws, and atomics there serialize on a handful of cachelines (the
    # measured hybrid cliff: floor 0 runs 65M words/s).  Round-2 final
    # sweep at vocab 1M / 400M words (2 runs each): floor 16 -> 0.899
    # planted-NN at 157M words/s (within 0.03 of full atomic's 0.927 at
    # 2.1x its speed); 32 -> 0.882 @ 165M; 64 -> 0.861 @ 171M
    # (benchmarks/results.md).  Default 16: the quality-defensible point;
    # raise it for more speed.
    hybrid_skip_rows: int = 16
    # Deprecated alias (round-1 API): True -> "atomic", False -> "hogwild".
    # None (default) leaves update_mode in charge.
    atomic_updates: Optional[bool] = None
    # --- multi-GPU engine --------------------------------------
    # "auto": fused at world 1; dp while tables are small; row-sharded
    # beyond (choose_engine above).
    # "dim": dimension-sharded (CIKM scheme, RCCL allreduce of partial dots).
    # "row": row-sharded parameter-server shape (RCCL alltoallv pull/push).
    # "dp": replicated tables + periodic delta-allreduce (fused kernel per
    # GPU; for vocabularies whose tables fit comfortably in HBM).
    engine: str = "auto"             # "auto" | "fused" | "dim" | "row" | "dp"
    chunk_words: int = 1 << 20       # dim-sharded feedback chunk
    f_correction: bool = True        # dim-sharded local-drift freshening
    sync_every: int = 4              # dp engine: steps between delta merges
    # mid-training checkpoints every N steps (0 = off; the reference has
    # none — saves only at the end, mllib:493-498).  Written next to the
    # final save_path as "<save_path>-step<N>"; each is a complete
    # loadable model directory.
    checkpoint_every: int = 0
    # --- semantics switches ---------------------
    # The reference's subsampling is a de-facto no-op (integer-division bug,
    # mllib:375-377).  We implement the intended math; set
    # legacy_subsample=True to reproduce the reference behaviour (keep all).
    legacy_subsample: bool = False
    # Reference windows are asymmetric b-left/(b-1)-right with possible empty
    # context (mllib:385-387).  Default is the canonical symmetric shrunk
    # window; "reference" reproduces B2.
    window_mode: str = "canonical"   # "canonical" | "reference"
    # HogBatch-style shared negative sampling (OPT-IN; changes the draw
    # layout, rng.py): one negative set per center position, reused
    # across its contexts.  Cuts per-position target-row traffic from
    # ~2b*(1+n) to ~2b+n rows — the only route past the measured HBM
    # floor at dim=1024 neg=25 (profiles/round2_kernel_stats.md);
    # quality measured on the planted-synonym probe
    # (benchmarks/results.md).  Default off: reference draw semantics.
    shared_negatives: bool = False
    # sigmoid: "exact" clipped sigmoid, or "lut" = the reference's
    # EXP_TABLE_SIZE-entry lookup (createExpTable/getSigmoid, mllib:281-302)
    sigmoid_mode: str = "exact"      # "exact" | "lut"

    def __post_init__(self) -> None:
        if self.atomic_updates is not None:
            self.update_mode = "atomic" if self.atomic_updates else "hogwild"
        self.validate()

    def resolved_update_mode(self) -> str:
        """update_mode, with the deprecated atomic_updates alias (possibly
        assigned after construction) taking precedence when set."""
        if self.atomic_updates is not None:
            return "atomic" if self.atomic_updates else "hogwild"
        return self.update_mode

    def effective_atomic_below(self) -> int:
        """Row-id threshold below which updates use atomics (the engines'
        kernel argument): 0 = hogwild everywhere, 2^31-1 = everywhere."""
        mode = self.resolved_update_mode()
        if mode == "hogwild":
            return 0
        if mode == "atomic":
            return 2 ** 31 - 1
        return int(self.hybrid_hot_rows)

    def effective_atomic_floor(self, vocab_size: "int | None" = None
                                ) -> int:
        # Rows below this stay hogwild even in atomic ranges (the hybrid
        # ultra-head contention escape; 0 elsewhere).  The escape exists
        # for million-row Zipf heads where one row absorbs ~1% of all
        # negative draws and serializes its cacheline; on small
        # vocabularies those same top rows ARE the content words (a 3.6k
        # vocabulary's "berlin" sits in the top 128), so the floor scales
        # down with vocabulary size.
        if self.resolved_update_mode() != "hybrid":
            return 0
        floor = int(self.hybrid_skip_rows)
        if vocab_size is not None:
            floor = min(floor, int(vocab_size) // 1024)
        return floor

    def validate(self) -> None:
        if self.vector_size <= 0:
            raise ValueError("vector_size must be > 0")
        if self.window <= 0:
            raise ValueError("window must be > 0")
        if self.n < 0:
            raise ValueError("n (negatives) must be >= 0")
        if self.batch_size <= 0:
            raise ValueError("batch_size must be > 0")
        if self.num_iterations <= 0:
            raise ValueError("num_iterations must be > 0")
        if self.min_count < 0:
            raise ValueError("min_count must be >= 0")
        if self.max_sentence_length <= 0:
            raise ValueError("max_sentence_length must be > 0")
        if self.unigram_table_size <= 0:
            raise ValueError("unigram_table_size must be > 0")
        if self.dtype not in ("auto", "float32", "bfloat16"):
            raise ValueError(f"unsupported dtype {self.dtype!r}")
        if self.update_mode not in ("hogwild", "atomic", "hybrid"):
            raise ValueError(f"unsupported update_mode {self.update_mode!r}")
        if self.hybrid_hot_rows < 0:
            raise ValueError("hybrid_hot_rows must be >= 0")
        if self.hybrid_skip_rows < 0:
            raise ValueError("hybrid_skip_rows must be >= 0")
        if self.window_mode not in ("canonical", "reference"):
            raise ValueError(f"unsupported window_mode {self.window_mode!r}")
        if self.engine not in ("auto", "fused", "dim", "row", "dp"):
            raise ValueError(f"unsupported engine {self.engine!r}")
        if self.sigmoid_mode not in ("exact", "lut"):
            raise ValueError(f"unsupported sigmoid_mode {self.sigmoid_mode!r}")
        if self.checkpoint_every < 0:
            raise ValueError("checkpoint_every must be >= 0")

    # -- (de)serialisation used by the checkpoint metadata ------------------
    def to_dict(self) -> dict:
        return dataclasses.asdict(self)

    @classmethod
    def from_dict(cls, d: dict) -> "Word2VecConfig":
        known = {f.name for f in dataclasses.fields(cls)}
        return cls(**{k: v for k, v in d.items() if k in known})

    def to_json(self) -> str:
        return json.dumps(self.to_dict(), indent=2, sort_keys=True)

    @classmethod
    def from_json(cls, s: str) -> "Word2VecConfig":
        return cls.from_dict(json.loads(s))
