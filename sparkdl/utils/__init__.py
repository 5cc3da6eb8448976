"""sparkdl.utils — observability and checkpoint helpers
(SURVEY.md §5 aux subsystems)."""

from sparkdl.utils.profiling import StepTimer, CommTimer  # noqa: F401
from sparkdl.utils.checkpoint import save_checkpoint, load_checkpoint  # noqa: F401
