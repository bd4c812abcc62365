from .optimizer import (Optimizer, SGD, NAG, Adam, AdamW, LAMB, RMSProp,
                        AdaGrad, Ftrl, Signum, create, register, Updater,
                        get_updater)
