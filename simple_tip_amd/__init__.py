"""simple_tip_amd — an MI355X-native DNN test-input-prioritization (TIP) and
active-learning engine.

Re-implements the capabilities of the ISSTA'22 reproduction package
``testingautomated-usi/simple-tip`` (reference layout: ``reproduction.py`` phase
CLI, ``/assets`` artifact fabric, prioritizer/APFD API), designed from scratch
for AMD Instinct MI355X (gfx950):

- PyTorch-ROCm models with activation-trace taps fused into the forward pass.
- Hand-written HIP/CDNA4 kernels (``simple_tip_amd/ops/hip``) for the hot
  numeric paths: MFMA pairwise-distance GEMM (DSA/KDE/Mahalanobis/kmeans),
  coverage bitmap profiling (NAC/KMNC/NBC/SNAC/TKNC), device-resident CAM
  greedy set cover, fused softmax-uncertainty epilogues.
- RCCL over xGMI (via ``torch.distributed``) for data-parallel sharding of
  test inputs and training ATs across the 8 GPUs of one node.
"""

__version__ = "0.1.0"
