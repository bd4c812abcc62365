from .amp import init, init_trainer, convert_model, scale_loss, unscale
from .loss_scaler import LossScaler
