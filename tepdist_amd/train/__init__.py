from tepdist_amd.train.optim import AdamW  # noqa: F401
from tepdist_amd.train.trainer import Trainer  # noqa: F401
