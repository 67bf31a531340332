from .spec import ModelSpec, get_spec  # noqa: F401
from .model import CausalLM  # noqa: F401
