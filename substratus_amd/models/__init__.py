from .config import ModelConfig, get_config, register
from .model import CausalLM
