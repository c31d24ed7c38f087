"""The consensus-polishing network, checkpoint-compatible with the reference.

Architecture (reference: roko/rnn_model.py:24-59, analysed in SURVEY.md §2.3):

  ids (B, 200, 90) int in [0, 12)
    -> Embedding(12, 50) + dropout                  (B, 200, 90, 50)
    -> permute: reads become the reduced axis       (B, 90, 50, 200)
    -> fc1 Linear(200 -> 100) + ReLU + dropout      (B, 90, 50, 100)
    -> fc2 Linear(100 -> 10)  + ReLU + dropout      (B, 90, 50, 10)
    -> reshape                                      (B, 90, 500)
    -> GRU(500, 128, 3 layers, bidirectional)       (B, 90, 256)
    -> fc4 Linear(256 -> 5)                         (B, 90, 5) logits

Parameter names (``embedding.weight``, ``fc1.*``, ``fc2.*``, ``gru.weight_ih_l*``
/ ``_reverse`` …, ``fc4.*``) form the ``.pth`` state_dict contract
(SURVEY.md §5.4): reference checkpoints load into this module unchanged.

On CUDA (= ROCm) devices the forward runs through hand-written CDNA4 HIP
kernels (roko_amd.ops); on CPU it runs the plain PyTorch ops and doubles as
the numerics reference for the kernel tests.
"""

from __future__ import annotations

import torch
from torch import nn

from . import config as C


def _init_gru(gru: nn.GRU) -> None:
    """Orthogonal init for matrices, small-normal for biases.

    Mirrors the reference's init policy (rnn_model.py:15-21): every parameter
    tensor with >= 2 dims is orthogonally initialised, 1-D params (biases)
    are drawn from N(0, 1) — the reference uses ``init.normal_`` defaults.
    """
    for name, p in gru.named_parameters():
        if p.dim() >= 2:
            nn.init.orthogonal_(p)
        else:
            nn.init.normal_(p)


class RokoModel(nn.Module):
    """Bidirectional-GRU consensus polisher (the reference's ``RNN``)."""

    def __init__(
        self,
        in_size: int = C.IN_SIZE,
        hidden_size: int = C.HIDDEN_SIZE,
        num_layers: int = C.NUM_LAYERS,
        dropout: float = C.DROPOUT,
    ) -> None:
        super().__init__()
        if in_size != C.EMBED_DIM * C.FC2_OUT:
            raise ValueError("in_size must equal EMBED_DIM * FC2_OUT")
        self.in_size = in_size
        self.hidden_size = hidden_size
        self.num_layers = num_layers

        self.embedding = nn.Embedding(C.NUM_BASE_IDS, C.EMBED_DIM)
        self.fc1 = nn.Linear(C.WINDOW_ROWS, C.FC1_OUT)
        self.fc2 = nn.Linear(C.FC1_OUT, C.FC2_OUT)
        self.gru = nn.GRU(
            in_size,
            hidden_size,
            num_layers=num_layers,
            batch_first=True,
            bidirectional=True,
            dropout=dropout,
        )
        _init_gru(self.gru)
        self.fc4 = nn.Linear(2 * hidden_size, C.NUM_CLASSES)
        self.dropout = nn.Dropout(dropout)

    # -- reference (CPU / autograd) path ------------------------------------
    def _forward_torch(self, x: torch.Tensor) -> torch.Tensor:
        e = self.dropout(self.embedding(x))          # (B, R, W, E)
        e = e.permute(0, 2, 3, 1)                    # (B, W, E, R)
        t = self.dropout(torch.relu(self.fc1(e)))    # (B, W, E, 100)
        t = self.dropout(torch.relu(self.fc2(t)))    # (B, W, E, 10)
        t = t.reshape(t.shape[0], t.shape[1], -1)    # (B, W, 500)
        out, _ = self.gru(t)                         # (B, W, 2H)
        return self.fc4(out)                         # (B, W, 5)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.dtype not in (torch.int64, torch.int32, torch.uint8):
            raise TypeError(f"expected integer base ids, got {x.dtype}")
        if x.dim() != 3 or x.shape[1] != C.WINDOW_ROWS or x.shape[2] != C.WINDOW_COLS:
            raise ValueError(f"expected (B, {C.WINDOW_ROWS}, {C.WINDOW_COLS}), got {tuple(x.shape)}")
        if x.is_cuda:
            from . import ops

            if ops.available():
                return ops.model_forward(self, x)
            ops.require()  # raises: HIP extension must be present on GPU hosts
        return self._forward_torch(x.long())

    # -- checkpoint helpers ---------------------------------------------------
    def load_reference_checkpoint(self, path: str, map_location="cpu") -> None:
        """Load a reference-format ``.pth`` state_dict."""
        state = torch.load(path, map_location=map_location, weights_only=True)
        self.load_state_dict(state)


def new_model() -> RokoModel:
    return RokoModel()
