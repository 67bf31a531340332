from .learner import Learner  # noqa: F401
from .optim import Adam8bit, make_optimizer  # noqa: F401
