"""AWD-LSTM (ULMFiT) language model, MI355X-native.

Re-creates the model that the reference trains and serves:
  * reference config surface: /root/reference/Issue_Embeddings/train.py:42-73
    (emb_sz, n_hid, n_layers, output_p, hidden_p, input_p, embed_p, weight_p,
    tie_weights, out_bias)
  * deployed shape: emb_sz=800, n_hid=2400, n_layers=4
    (/root/reference/Issue_Embeddings/notebooks/04_Inference.ipynb:56-60)

Checkpoint compatibility: parameter/buffer names reproduce the fastai-1.x
``AWD_LSTM`` state-dict layout (``encoder.weight``, ``encoder_dp.emb.weight``,
``rnns.{l}.weight_hh_l0_raw``, ``rnns.{l}.module.weight_ih_l0``,
``rnns.{l}.module.weight_hh_l0``, ``rnns.{l}.module.bias_{ih,hh}_l0``) so that
``learn.save_encoder()`` artifacts load directly (SURVEY.md §5 "checkpoint").

The compute path is NOT a port of fastai: on ROCm devices every LSTM layer
runs through the fused CDNA4 HIP cell kernel (ops/lstm.py); the CPU path is a
plain PyTorch reference implementation used for tests and small smoke runs.
"""
from __future__ import annotations

import math
from typing import List, Optional, Tuple

import torch
from torch import Tensor, nn

from ..ops.lstm import lstm_forward
from ..ops.dropout import variational_dropout, dropconnect
from ..ops.embedding import embedding_row_dropout

__all__ = [
    "EmbeddingDropout",
    "WeightDroppedLSTM",
    "WeightDroppedQRNN",
    "RNNDropout",
    "AWDLSTMEncoder",
    "LinearDecoder",
    "AWDLSTM",
    "awd_lstm_lm_config",
]

# fastai's awd_lstm_lm_config defaults, as mutated by the reference
# (train.py:68-73): output_p .1, hidden_p .15, input_p .25, embed_p .02,
# weight_p .2, tie_weights True, out_bias True.
awd_lstm_lm_config = dict(
    emb_sz=400,
    n_hid=1150,
    n_layers=3,
    pad_token=1,
    qrnn=False,
    bidir=False,
    tie_weights=True,
    out_bias=True,
    output_p=0.1,
    hidden_p=0.15,
    input_p=0.25,
    embed_p=0.02,
    weight_p=0.2,
)


class RNNDropout(nn.Module):
    """Variational (locked) dropout: one (B,1,H) mask broadcast over time.

    Reference semantics: fastai RNNDropout used between LSTM layers
    (hidden_p) and on the embedding output (input_p) — SURVEY.md §2.4 K4.
    """

    def __init__(self, p: float = 0.5):
        super().__init__()
        self.p = p

    def forward(self, x: Tensor) -> Tensor:
        return variational_dropout(x, self.p, self.training)

    def extra_repr(self) -> str:
        return f"p={self.p}"


class EmbeddingDropout(nn.Module):
    """Word-level embedding dropout: zero whole rows of the table (K1).

    Matches fastai EmbeddingDropout: a (vocab,1) Bernoulli mask scaled by
    1/(1-p) applied to the weight before lookup, so every occurrence of a
    dropped word is zeroed consistently within a batch.
    """

    def __init__(self, emb: nn.Embedding, embed_p: float):
        super().__init__()
        self.emb = emb
        self.embed_p = embed_p

    def forward(self, words: Tensor, scale: Optional[float] = None) -> Tensor:
        if words.is_cuda:
            # K1 HIP kernel: row mask fused into the gather — the masked
            # 60kx800 table is never materialized (embedding.hip)
            return embedding_row_dropout(
                self.emb.weight, words, self.embed_p, self.training,
                self.emb.padding_idx, scale)
        if self.training and self.embed_p != 0:
            size = (self.emb.weight.size(0), 1)
            mask = self.emb.weight.new_empty(size).bernoulli_(1 - self.embed_p)
            mask = mask / (1 - self.embed_p)
            masked_embed = self.emb.weight * mask
        else:
            masked_embed = self.emb.weight
        if scale is not None:
            masked_embed = masked_embed * scale
        pad_idx = self.emb.padding_idx
        if pad_idx is None:
            pad_idx = -1
        return nn.functional.embedding(
            words, masked_embed, pad_idx, self.emb.max_norm, self.emb.norm_type,
            self.emb.scale_grad_by_freq, self.emb.sparse)


class _LSTMParams(nn.Module):
    """Name shim so state-dict keys match fastai's ``rnns.{l}.module.*``.

    Holds the non-recurrent LSTM parameters. ``weight_hh_l0`` is kept as a
    (non-trainable) buffer mirroring the fastai hack of stashing the dropped
    weight back onto the inner nn.LSTM — present in reference .pth encoders.
    """

    def __init__(self, input_size: int, hidden_size: int):
        super().__init__()
        self.weight_ih_l0 = nn.Parameter(torch.empty(4 * hidden_size, input_size))
        self.bias_ih_l0 = nn.Parameter(torch.zeros(4 * hidden_size))
        self.bias_hh_l0 = nn.Parameter(torch.zeros(4 * hidden_size))
        # mirror of the raw recurrent weight; persisted for .pth parity only
        self.register_buffer("weight_hh_l0", torch.empty(4 * hidden_size, hidden_size))


class WeightDroppedLSTM(nn.Module):
    """Single LSTM layer with DropConnect on the recurrent weights (K2+K3).

    Gate layout follows PyTorch/cuDNN order (i, f, g, o) so fastai/PyTorch
    checkpoints load unchanged. The recurrent mask is re-sampled per forward
    in training (weight_p=0.2 in the deployed config, train.py:70).
    """

    def __init__(self, input_size: int, hidden_size: int, weight_p: float = 0.0):
        super().__init__()
        self.input_size, self.hidden_size, self.weight_p = input_size, hidden_size, weight_p
        self.weight_hh_l0_raw = nn.Parameter(torch.empty(4 * hidden_size, hidden_size))
        self.module = _LSTMParams(input_size, hidden_size)
        self.reset_parameters()

    def reset_parameters(self) -> None:
        stdv = 1.0 / math.sqrt(self.hidden_size)
        for w in (self.weight_hh_l0_raw, self.module.weight_ih_l0,
                  self.module.bias_ih_l0, self.module.bias_hh_l0):
            nn.init.uniform_(w, -stdv, stdv)
        with torch.no_grad():
            self.module.weight_hh_l0.copy_(self.weight_hh_l0_raw)

    def _masked_weight(self) -> Tensor:
        # K3: seeded DropConnect kernel on CUDA (no mask tensor, in-place
        # grad masking); F.dropout reference on CPU (ops/dropout.py)
        return dropconnect(self.weight_hh_l0_raw, self.weight_p,
                           self.training)

    def forward(self, x: Tensor, state: Tuple[Tensor, Tensor]) -> Tuple[Tensor, Tuple[Tensor, Tensor]]:
        """x: (B, T, input_size); state: (h, c) each (B, hidden_size)."""
        w_hh = self._masked_weight()
        out, (h, c) = lstm_forward(
            x, state[0], state[1], self.module.weight_ih_l0, w_hh,
            self.module.bias_ih_l0, self.module.bias_hh_l0)
        return out, (h, c)

    def _load_from_state_dict(self, state_dict, prefix, *args, **kwargs):
        # accept checkpoints missing the mirror buffer
        mirror = prefix + "module.weight_hh_l0"
        if mirror not in state_dict:
            state_dict[mirror] = state_dict.get(
                prefix + "weight_hh_l0_raw", self.module.weight_hh_l0).detach().clone()
        super()._load_from_state_dict(state_dict, prefix, *args, **kwargs)


class WeightDroppedQRNN(nn.Module):
    """Single QRNN layer (reference ``--qrnn`` flag, train.py:43).

    Gates for every timestep come from ONE hipBLASLt GEMM; the sequential
    part is only the elementwise fo-pool scan (ops/qrnn.py — HIP kernel on
    GPU). DropConnect (weight_p) applies to the gate projection like fastai
    wraps QRNNLayer.linear. window=2 keeps the previous window's last x
    (save_prev_x) so BPTT windows chain exactly.
    """

    def __init__(self, input_size: int, hidden_size: int,
                 weight_p: float = 0.0, window: int = 1):
        super().__init__()
        from ..ops.qrnn import qrnn_forward
        self._qrnn_forward = qrnn_forward
        self.input_size, self.hidden_size = input_size, hidden_size
        self.weight_p, self.window = weight_p, window
        self.weight_raw = nn.Parameter(
            torch.empty(3 * hidden_size, window * input_size))
        self.bias = nn.Parameter(torch.zeros(3 * hidden_size))
        self.prev_x: Optional[Tensor] = None
        stdv = 1.0 / math.sqrt(hidden_size)
        nn.init.uniform_(self.weight_raw, -stdv, stdv)

    def reset(self) -> None:
        self.prev_x = None

    def forward(self, x: Tensor, state: Tuple[Tensor, Tensor]
                ) -> Tuple[Tensor, Tuple[Tensor, Tensor]]:
        w = nn.functional.dropout(self.weight_raw, p=self.weight_p,
                                  training=self.training)
        prev = self.prev_x
        if prev is not None and (prev.size(0) != x.size(0)
                                 or prev.device != x.device):
            prev = None
        h, cT = self._qrnn_forward(x, state[1], w, self.bias,
                                   self.window, prev)
        if self.window == 2:
            nx = x[:, -1:].detach()
            if (self.prev_x is not None and self.prev_x.shape == nx.shape
                    and self.prev_x.device == nx.device
                    and self.prev_x.dtype == nx.dtype):
                # copy into the STABLE buffer instead of reassigning: a
                # fresh allocation here inside a hipGraph capture lives in
                # the capture-private pool and the python reassignment
                # frees the captured pointer -> memory fault on replay
                # (the r1 QRNN graph-capture fault, scripts/qrnn_graph_repro)
                self.prev_x.copy_(nx)
            else:
                self.prev_x = nx.clone()
        return h, (h[:, -1], cT)


class AWDLSTMEncoder(nn.Module):
    """The encoder stack: embedding (+dropout) → n_layers weight-dropped LSTMs.

    Equivalent of fastai ``AWD_LSTM`` (what ``learn.model[0]`` /
    ``save_encoder`` holds — reference inference.py:28-44 extracts it).
    Hidden state is carried across batches until ``reset()`` — the LSTM
    analogue of long-context streaming.
    """

    initrange = 0.1

    def __init__(self, vocab_sz: int, emb_sz: int, n_hid: int, n_layers: int,
                 pad_token: int = 1, hidden_p: float = 0.15, input_p: float = 0.25,
                 embed_p: float = 0.02, weight_p: float = 0.2,
                 qrnn: bool = False):
        super().__init__()
        self.vocab_sz, self.emb_sz, self.n_hid, self.n_layers = vocab_sz, emb_sz, n_hid, n_layers
        self.pad_token = pad_token
        self.qrnn = qrnn
        self.bs = 1
        self.encoder = nn.Embedding(vocab_sz, emb_sz, padding_idx=pad_token)
        self.encoder.weight.data.uniform_(-self.initrange, self.initrange)
        self.encoder_dp = EmbeddingDropout(self.encoder, embed_p)
        if qrnn:
            # reference --qrnn flag (train.py:43); fastai windows: 2 on the
            # first layer, 1 afterwards
            self.rnns = nn.ModuleList([
                WeightDroppedQRNN(
                    emb_sz if l == 0 else n_hid,
                    n_hid if l != n_layers - 1 else emb_sz,
                    weight_p=weight_p, window=2 if l == 0 else 1)
                for l in range(n_layers)
            ])
        else:
            self.rnns = nn.ModuleList([
                WeightDroppedLSTM(
                    emb_sz if l == 0 else n_hid,
                    n_hid if l != n_layers - 1 else emb_sz,
                    weight_p=weight_p)
                for l in range(n_layers)
            ])
        self.input_dp = RNNDropout(input_p)
        self.hidden_dps = nn.ModuleList([RNNDropout(hidden_p) for _ in range(n_layers)])
        self.hidden: List[Tuple[Tensor, Tensor]] = []
        self.reset()

    def _one_hidden(self, l: int, bs: int) -> Tuple[Tensor, Tensor]:
        nh = self.n_hid if l != self.n_layers - 1 else self.emb_sz
        p = self.encoder.weight
        return (p.new_zeros(bs, nh).detach(), p.new_zeros(bs, nh).detach())

    def reset(self, bs: Optional[int] = None) -> None:
        """Clear hidden state (between independent documents — inference.py:60,70)."""
        if bs is not None:
            self.bs = bs
        self.hidden = [self._one_hidden(l, self.bs) for l in range(self.n_layers)]
        for rnn in self.rnns:
            if hasattr(rnn, "reset"):
                rnn.reset()  # QRNN: clear the saved window-2 prev_x

    def select_hidden(self, idxs: Tensor) -> None:
        self.hidden = [(h[idxs], c[idxs]) for h, c in self.hidden]
        self.bs = len(idxs)

    def forward(self, input_ids: Tensor, from_embeddings: bool = False
                ) -> Tuple[List[Tensor], List[Tensor]]:
        """input_ids: (B, T) int64.  Returns (raw_outputs, dropped_outputs),
        one per layer — the fastai contract the pooling/decoder layers use."""
        bs = input_ids.size(0)
        if bs != self.bs:
            self.reset(bs)
        raw_output = self.input_dp(
            input_ids if from_embeddings else self.encoder_dp(input_ids))
        new_hidden, raw_outputs, outputs = [], [], []
        for l, (rnn, hid_dp) in enumerate(zip(self.rnns, self.hidden_dps)):
            raw_output, new_h = rnn(raw_output, self.hidden[l])
            new_hidden.append(new_h)
            raw_outputs.append(raw_output)
            if l != self.n_layers - 1:
                raw_output = hid_dp(raw_output)
            outputs.append(raw_output)
        self.hidden = [(h.detach(), c.detach()) for h, c in new_hidden]
        return raw_outputs, outputs


class LinearDecoder(nn.Module):
    """Tied-weight decoder: logits = h·Eᵀ + bias (K6). train.py:70 parity."""

    initrange = 0.1

    def __init__(self, n_out: int, n_hid: int, output_p: float,
                 tie_encoder: Optional[nn.Embedding] = None, bias: bool = True):
        super().__init__()
        self.decoder = nn.Linear(n_hid, n_out, bias=bias)
        self.decoder.weight.data.uniform_(-self.initrange, self.initrange)
        self.output_dp = RNNDropout(output_p)
        if bias:
            self.decoder.bias.data.zero_()
        if tie_encoder is not None:
            self.decoder.weight = tie_encoder.weight

    def forward(self, inputs: Tuple[List[Tensor], List[Tensor]]):
        raw_outputs, outputs = inputs
        output = self.output_dp(outputs[-1])
        decoded = self.decoder(output)
        return decoded, raw_outputs, outputs


class AWDLSTM(nn.Module):
    """Full language model: encoder (index 0) + decoder (index 1).

    Indexable like fastai's SequentialRNN so ``model[0]`` is the encoder
    (inference.py:44 does ``learn.model[0]``).
    """

    def __init__(self, vocab_sz: int, emb_sz: int = 400, n_hid: int = 1150,
                 n_layers: int = 3, pad_token: int = 1, tie_weights: bool = True,
                 out_bias: bool = True, output_p: float = 0.1, hidden_p: float = 0.15,
                 input_p: float = 0.25, embed_p: float = 0.02, weight_p: float = 0.2,
                 qrnn: bool = False, bidir: bool = False):
        super().__init__()
        if bidir:
            raise NotImplementedError(
                "bidir=True is not supported: the reference exposes the flag "
                "(train.py:43) but never enables it, and a bidirectional LM "
                "leaks future tokens into next-token prediction")
        encoder = AWDLSTMEncoder(vocab_sz, emb_sz, n_hid, n_layers, pad_token,
                                 hidden_p, input_p, embed_p, weight_p, qrnn=qrnn)
        decoder = LinearDecoder(vocab_sz, emb_sz, output_p,
                                tie_encoder=encoder.encoder if tie_weights else None,
                                bias=out_bias)
        self.layers = nn.ModuleList([encoder, decoder])

    def __getitem__(self, idx: int) -> nn.Module:
        return self.layers[idx]

    @property
    def encoder(self) -> AWDLSTMEncoder:
        return self.layers[0]

    @property
    def decoder(self) -> LinearDecoder:
        return self.layers[1]

    def reset(self, bs: Optional[int] = None) -> None:
        self.encoder.reset(bs)

    def forward(self, input_ids: Tensor):
        return self.layers[1](self.layers[0](input_ids))

    # --- fastai-compatible (de)serialization -------------------------------
    def save_encoder(self, path) -> None:
        """Save ``model[0]`` exactly like fastai ``learn.save_encoder``."""
        torch.save(self.encoder.state_dict(), path)

    def load_encoder(self, path, map_location="cpu") -> None:
        sd = torch.load(path, map_location=map_location, weights_only=True)
        if isinstance(sd, dict) and "model" in sd and isinstance(sd["model"], dict):
            sd = sd["model"]  # fastai learner .pth wraps under 'model'
        self.encoder.load_state_dict(sd)
