from .rnn_layer import RNN, LSTM, GRU
from .rnn_cell import (RNNCell, LSTMCell, GRUCell, SequentialRNNCell,
                       BidirectionalCell, DropoutCell, ZoneoutCell,
                       ResidualCell)
