from .basic import (
    BasePipeline,
    ImageClassificationPipeline,
    TextClassificationPipeline,
    TextGenerationPipeline,
)
from .generator import Generator

__all__ = [
    "BasePipeline",
    "TextGenerationPipeline",
    "TextClassificationPipeline",
    "ImageClassificationPipeline",
    "Generator",
]
