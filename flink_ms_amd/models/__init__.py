from .als import ALSConfig, ALSModel, ALSTrainer, train_als  # noqa: F401
from .svm import SVMConfig, SVMModel, SVMTrainer  # noqa: F401
from .mse import MSEResult, evaluate_mse  # noqa: F401
from .mean_vector import mean_vector_rows, mean_vector_tensor  # noqa: F401
from . import generator  # noqa: F401
