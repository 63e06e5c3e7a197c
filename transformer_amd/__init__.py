"""transformer_amd — MI355X-native encoder-decoder Transformer training framework.

A from-scratch reimplementation of the capability surface of
kuetuofa/Transformer (TF2 seq2seq NMT Transformer) designed
MI355X-first: PyTorch-ROCm is the tensor/autograd ledger, all hot device math
runs in hand-written CDNA4 (gfx950) HIP kernels on MFMA with LDS-staged tiles,
and data-parallel training uses RCCL over xGMI with bucketed all-reduce
overlapped with backward.

Layout:
    models/    Transformer / Encoder / Decoder (ref: Transformer.py, Encoder.py,
               Decoder.py in the reference)
    ops/       hand-written HIP kernel library + autograd wrappers + the pure
               PyTorch fp32 reference ops used as CPU path and test oracle
    parallel/  bucketed gradient all-reduce DP runtime (ref:
               distributed_train.py's MirroredStrategy)
    data/      text pipeline + subword tokenizer + synthetic benchmark data
               (ref: utils.py)
    runtime/   Train/DistributedTrain loops, Noam schedule, checkpointing,
               metrics, TensorBoard-format summaries (ref: train.py)
"""

__version__ = "0.1.0"
