"""AMP op lists (parity with python/mxnet/contrib/amp/lists/symbol_fp16.py,
re-targeted bf16-first for CDNA4: bf16 MFMA is the fast path; fp32 is kept
for numerically-sensitive reductions)."""

# ops that run in low precision (bf16) — MFMA/bandwidth-bound
LP16_FUNCS = [
    "FullyConnected", "Convolution", "dot", "batch_dot",
    "multi_head_attention", "Activation", "relu", "sigmoid", "tanh",
    "gelu", "silu", "Pooling", "Embedding",
]

# ops forced to fp32 (reductions / losses / normalization statistics)
FP32_FUNCS = [
    "softmax", "log_softmax", "SoftmaxOutput", "softmax_cross_entropy",
    "norm", "mean", "sum", "BatchNorm", "LayerNorm", "InstanceNorm",
    "GroupNorm", "smooth_l1", "exp", "log",
]

# ops that take the widest input type among their args
WIDEST_TYPE_CASTS = [
    "elemwise_add", "elemwise_mul", "elemwise_sub", "elemwise_div",
    "broadcast_add", "broadcast_mul", "broadcast_sub", "broadcast_div",
    "add_n", "concat", "stack", "where",
]
