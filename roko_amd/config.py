"""Central configuration: every constant that defines an interop contract.

These constants freeze the three contracts the framework exposes (mirroring the
reference's hard-coded module constants):

1. the feature-window geometry (reference: include/generate.h:19-23,
   roko/features.py:11-16),
2. the base / label encodings (reference: generate.cpp:17-25, roko/labels.py:6-10),
3. the model shape whose parameter names form the ``.pth`` checkpoint contract
   (reference: roko/rnn_model.py:10-12,24-44).

Nothing else in the package may re-define these numbers.
"""

from dataclasses import dataclass

# ---------------------------------------------------------------------------
# Window geometry (reference: include/generate.h:19-23, features.py:16)
# ---------------------------------------------------------------------------
#: rows per feature window = number of sampled read rows
WINDOW_ROWS = 200
#: columns per feature window = (position, insertion) slots covered
WINDOW_COLS = 90
#: stride between consecutive windows, in pileup columns. Consecutive windows
#: overlap by WINDOW_COLS - WINDOW_STRIDE = 60 columns, so every column is
#: covered by up to 3 windows (the basis of the inference majority vote).
WINDOW_STRIDE = 30
#: maximum insertion slots materialised after a reference position
MAX_INS = 3
#: contigs are split into regions of this many bases for process parallelism
#: (reference: features.py:16)
REGION_SIZE = 100_000
#: neighbouring regions overlap by this many bases so that window joins at
#: region boundaries are covered (reference: features.py:16-27)
REGION_OVERLAP = 300

# ---------------------------------------------------------------------------
# Base / label encodings (reference: generate.cpp:17-25, labels.py:6-10)
# ---------------------------------------------------------------------------
#: feature alphabet: A C G T GAP UNKNOWN, + STRAND_OFFSET if the read is
#: reverse-complemented => ids 0..11 feed a 12-row embedding table
BASE_A, BASE_C, BASE_G, BASE_T, BASE_GAP, BASE_UNKNOWN = 0, 1, 2, 3, 4, 5
STRAND_OFFSET = 6
NUM_BASE_IDS = 12

#: label alphabet (truth classes). '*' = deletion w.r.t. the draft.
LABEL_ALPHABET = "ACGT*N"
LABEL_ENCODING = {b: i for i, b in enumerate(LABEL_ALPHABET)}
LABEL_DECODING = {i: b for i, b in enumerate(LABEL_ALPHABET)}
LABEL_GAP = 4  # '*'
LABEL_UNKNOWN = 5  # 'N' — windows containing one are dropped before training
#: the classifier head is 5-way: UNKNOWN-labelled windows never reach training
NUM_CLASSES = 5

# ---------------------------------------------------------------------------
# Read filtering (reference: models.cpp:13-33, models.h:22-23)
# ---------------------------------------------------------------------------
#: BAM flag mask of reads dropped from the pileup: UNMAP | SECONDARY | QCFAIL
#: | DUP | SUPPLEMENTARY
FILTER_FLAG = 0x4 | 0x100 | 0x200 | 0x400 | 0x800
MIN_MAPQ = 10

# ---------------------------------------------------------------------------
# Model shape (reference: rnn_model.py:10-12). These define the .pth contract.
# ---------------------------------------------------------------------------
EMBED_DIM = 50
FC1_OUT = 100
FC2_OUT = 10
#: GRU input size = EMBED_DIM * FC2_OUT after the per-column reduction
IN_SIZE = EMBED_DIM * FC2_OUT  # 500
HIDDEN_SIZE = 128
NUM_LAYERS = 3
DROPOUT = 0.2

#: state_dict keys the checkpoint contract requires (reference:
#: rnn_model.py:28-44 module names)
CHECKPOINT_KEYS = (
    ["embedding.weight", "fc1.weight", "fc1.bias", "fc2.weight", "fc2.bias",
     "fc4.weight", "fc4.bias"]
    + [f"gru.{kind}_l{layer}{suffix}"
       for layer in range(NUM_LAYERS)
       for suffix in ("", "_reverse")
       for kind in ("weight_ih", "weight_hh", "bias_ih", "bias_hh")]
)

# ---------------------------------------------------------------------------
# Training hyper-parameters (reference: train.py:12-15)
# ---------------------------------------------------------------------------
BATCH_SIZE = 128
EPOCHS = 100
LR = 1e-4
PATIENCE = 7


@dataclass
class FeatureConfig:
    """Runtime-tunable knobs of feature generation."""

    window_rows: int = WINDOW_ROWS
    window_cols: int = WINDOW_COLS
    window_stride: int = WINDOW_STRIDE
    max_ins: int = MAX_INS
    region_size: int = REGION_SIZE
    region_overlap: int = REGION_OVERLAP
    filter_flag: int = FILTER_FLAG
    min_mapq: int = MIN_MAPQ
    #: RNG seed for read-row sampling; each region derives its own stream from
    #: (seed, contig, region start) so results are reproducible regardless of
    #: worker scheduling (the reference re-seeds srand(time) per call instead,
    #: gen.cpp:12 — deliberately fixed here).
    seed: int = 0


@dataclass
class TrainConfig:
    batch_size: int = BATCH_SIZE
    epochs: int = EPOCHS
    lr: float = LR
    patience: int = PATIENCE
    workers: int = 0
    in_memory: bool = False
    seed: int = 0
    #: resume from <out>/train_state.pt when present
    resume: bool = False
    #: gradient-bucket size (bytes) for the RCCL all-reduce overlap
    bucket_bytes: int = 2 << 20


MODEL_DTYPES = ("float32", "bfloat16")
