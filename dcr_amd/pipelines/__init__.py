from .text2img import StableDiffusionPipeline, Newpipe

__all__ = ["StableDiffusionPipeline", "Newpipe"]
