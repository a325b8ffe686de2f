from automodel_amd.speculative.decode import (  # noqa: F401
    EagleProposer,
    NgramProposer,
    SpecStats,
    speculative_generate,
)
from automodel_amd.speculative.draft import EagleDraftConfig, EagleDraftModel  # noqa: F401
