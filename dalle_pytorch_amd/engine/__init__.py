from dalle_pytorch_amd.engine.decode import FastDecoder

__all__ = ['FastDecoder']
