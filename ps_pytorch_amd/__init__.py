"""ps_pytorch_amd — an MI355X-native synchronous parameter-server training framework.

A from-scratch re-design of the capabilities of hwang595/ps_pytorch for AMD
Instinct MI355X (gfx950, CDNA4): PyTorch-ROCm for autograd, hand-written HIP
kernels for the hot ops (fused PS update, gradient wire pack/unpack, fused
conv/BN/ReLU & MFMA GEMM), and RCCL collectives over xGMI for transport
(1 PS + N workers on one 8-GPU node, `torch.distributed` backend "nccl").

Layer map (see SURVEY.md for the reference blueprint this re-implements):
  models/    LeNet / ResNet / VGG families      (ref: src/model_ops/)
  parallel/  flat param/grad space, RCCL transport, PS/worker/evaluator roles
             (ref: sync_replicas_master_nn.py, distributed_worker.py,
              distributed_evaluator.py, data_parallel_dist/)
  optim/     PS-side SGD/Adam on the flat buffer (ref: src/optim/)
  ops/       HIP/CDNA4 kernels + loaders        (ref: compression.py + the
             torch C++ ops the reference drives from Python)
  data/      synthetic GPU-resident datasets    (ref: src/util.py prepare_data,
             src/data_loader_ops/my_data_loader.py)
  utils/     logging / checkpoint / metrics     (ref: log-line + NFS layout)
"""

__version__ = "0.1.0"
