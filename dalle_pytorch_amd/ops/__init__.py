"""First-party op library.

Every hot op from the reference inventory routes through
here. On an AMD GPU the CDNA4 HIP extension (``dalle_pytorch_amd/_hip*.so``,
built in-tree for gfx950) is mandatory — ops raise if it is missing rather
than silently falling back to eager PyTorch. On CPU the eager oracle path
runs, which is also what the numerics tests compare the kernels against.
"""

from dalle_pytorch_amd.ops.dispatch import hip_available, hip_module, using_eager_fallback
from dalle_pytorch_amd.ops.attention import attention_core, axial_attention
from dalle_pytorch_amd.ops.fused import geglu

__all__ = ['attention_core', 'axial_attention', 'geglu', 'hip_available', 'hip_module', 'using_eager_fallback']
