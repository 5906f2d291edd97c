"""dalle_pytorch_amd — an MI355X-native DALL-E training/generation framework.

Public surface mirrors lucidrains/DALLE-pytorch (`dalle_pytorch/__init__.py:1-5`):
the five model classes plus ``__version__``. Everything underneath is a fresh
CDNA4/HIP + RCCL design, not a port.
"""

from dalle_pytorch_amd.version import __version__
from dalle_pytorch_amd.models.dvae import DiscreteVAE
from dalle_pytorch_amd.models.dalle import DALLE
from dalle_pytorch_amd.models.clip import CLIP
from dalle_pytorch_amd.models.vae_adapters import OpenAIDiscreteVAE, VQGanVAE

__all__ = ['DALLE', 'CLIP', 'DiscreteVAE', 'OpenAIDiscreteVAE', 'VQGanVAE', '__version__']
