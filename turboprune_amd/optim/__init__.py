from turboprune_amd.optim.sgd import FusedMaskedSGD, ScheduleFreeSGD  # noqa: F401
from turboprune_amd.optim import schedulers  # noqa: F401
from turboprune_amd.optim.schedulers import build_scheduler  # noqa: F401
from turboprune_amd.optim.sgd import build_optimizer  # noqa: F401
