from turboprune_amd.pruning.prune import (  # noqa: F401
    prune_the_model,
    prune_mag,
    prune_snip,
    prune_synflow,
    prune_random_erk,
    prune_random_balanced,
    prune_er_erk,
    prune_er_balanced,
    erk_keep_probabilities,
    balanced_keep_probabilities,
)
