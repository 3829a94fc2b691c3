"""distegnn_amd — an MI355X-native FastEGNN/DistEGNN training framework.

A from-scratch rebuild of the capabilities of the reference DistEGNN repo
(GLAD-RUC/DistEGNN) designed for AMD Instinct MI355X (gfx950):

* PyTorch-ROCm for the module system / autograd,
* hand-written HIP/CDNA4 kernels (``distegnn_amd/ops/csrc``) for the hot
  message-passing ops (segment reductions, graph pooling, radius graph,
  fused MFMA edge/virtual-edge blocks),
* RCCL (``torch.distributed`` backend "nccl" on ROCm) over xGMI for the
  graph-partition spatial parallelism ("DistEGNN") and gradient sync.

Layout
------
``utils``     config containers, seeding, rotations
``data``      Data/Batch graph containers, loaders, synthetic generators,
              offline preprocessing + graph partitioners
``ops``       op dispatch: HIP extension on GPU, eager fp32 reference on CPU
``parallel``  RCCL process-group setup, differentiable fused collectives,
              flat gradient bucket
``models``    FastEGNN (primary), FastRF, FastSchNet, SchNet, EGNN, RF,
              Linear baselines
``runtime``   training loop, loss assembly (coord MSE + MMD), checkpointing
"""

__version__ = "0.1.0"
