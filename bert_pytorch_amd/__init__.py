"""bert-pytorch_amd — an MI355X-native BERT/RoBERTa pretraining framework.

A from-scratch re-design of the capabilities of gpauloski/BERT-PyTorch for
AMD Instinct MI355X (gfx950 / CDNA4): PyTorch-ROCm is the framework layer,
every performance-critical op the reference outsourced to NVIDIA Apex /
amp_C / Rust tokenizers is an in-repo HIP (gfx950) or C++ component, and
data parallelism runs on RCCL over xGMI.

Layout:
    models/    BERT encoder + task heads (reference: src/modeling.py)
    ops/       hand-written HIP kernel wrappers + eager fp32 references
    optim/     fused LAMB/Adam/BertAdam + LR schedulers + K-FAC
               (reference: src/optimization.py, src/schedulers.py)
    data/      sharded HDF5 dataset, dynamic masking, chunked sampler,
               in-repo minimal HDF5 I/O (reference: src/dataset.py)
    parallel/  torch.distributed/RCCL helpers, DDP wrapping tuned for xGMI
    utils/     logging sinks, checkpoint I/O, profiling helpers
"""

__version__ = "0.1.0"
