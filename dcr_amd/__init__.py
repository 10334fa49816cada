"""dcr_amd — an MI355X-native diffusion-replication research framework.

A from-scratch rebuild of the capabilities of somepago/DCR (the CVPR'23
"Diffusion Art or Digital Forgery?" / "Understanding and Mitigating Copying
in Diffusion Models" codebase) designed MI355X-first:

* PyTorch-ROCm as the tensor/autograd frontend,
* hand-written HIP/CDNA4 (gfx950) kernels for the fused hot ops
  (GroupNorm+SiLU, LayerNorm, GEGLU, flash attention, fused AdamW,
  diffusion-scheduler math) — see ``dcr_amd/ops``,
* RCCL over xGMI for every collective (``torch.distributed`` backend
  "nccl" on ROCm), one process per GPU,
* diffusers-compatible checkpoint layout (``checkpoint_{step}/unet/...``)
  without depending on diffusers at runtime.

Reference capability map: /root/repo/SURVEY.md.
"""

__version__ = "0.1.0"
