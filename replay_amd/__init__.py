"""replay_amd — an MI355X-native recommender-systems framework.

A from-scratch framework with the capabilities of sb-ai-lab/RePlay, designed
for AMD Instinct MI355X (gfx950/CDNA4): PyTorch-ROCm training loops with
hand-written HIP kernels on the hot path and RCCL collectives over xGMI for
multi-GPU data-parallel / negative-sharing / catalog-sharded execution.
"""

__version__ = "0.1.0"
