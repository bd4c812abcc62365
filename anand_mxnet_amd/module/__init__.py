from .base_module import BaseModule
from .module import Module
