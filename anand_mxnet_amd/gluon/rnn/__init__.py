from .rnn_cell import RNNCell, LSTMCell, GRUCell, SequentialRNNCell
from .rnn_layer import RNN, LSTM, GRU
