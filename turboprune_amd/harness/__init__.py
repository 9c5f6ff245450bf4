from turboprune_amd.harness.base import BaseHarness  # noqa: F401
from turboprune_amd.harness.pruning import PruningHarness  # noqa: F401
from turboprune_amd.harness.cyclic import CyclicPruningHarness  # noqa: F401
