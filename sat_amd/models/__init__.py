from .caption_generator import CaptionGenerator
from .base_model import BaseModel
