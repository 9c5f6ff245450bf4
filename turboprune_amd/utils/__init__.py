from turboprune_amd.utils.experiment import (  # noqa: F401
    gen_expt_dir,
    set_seed,
    generate_densities,
    generate_cyclical_schedule,
    save_model,
    resume_experiment,
)
from turboprune_amd.utils.logging import MetricsLogger, Throughput  # noqa: F401
from turboprune_amd.utils.console import (  # noqa: F401
    display_training_info,
    reset_optimizer,
)
