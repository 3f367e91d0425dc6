from .grad_averager import GradientAverager
from .grad_scaler import GradScaler
from .optimizer import Optimizer
from .progress_tracker import GlobalTrainingProgress, LocalTrainingProgress, ProgressTracker
from .state_averager import TrainingStateAverager
from .power_sgd_averager import PowerSGDGradientAverager
from .training_averager import TrainingAverager
