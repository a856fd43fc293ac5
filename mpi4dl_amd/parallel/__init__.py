from .partition import model_generator  # noqa: F401
from .pipeline import train_model  # noqa: F401
