from . import vision
from .vision import get_model
