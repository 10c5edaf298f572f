from .awd_lstm import (AWDLSTM, AWDLSTMEncoder, EmbeddingDropout,
                       LinearDecoder, RNNDropout, WeightDroppedLSTM,
                       WeightDroppedQRNN, awd_lstm_lm_config)

__all__ = [
    "AWDLSTM", "AWDLSTMEncoder", "EmbeddingDropout", "LinearDecoder",
    "RNNDropout", "WeightDroppedLSTM", "WeightDroppedQRNN",
    "awd_lstm_lm_config",
]
