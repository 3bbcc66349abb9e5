from dlrover_amd.unified.api import DLJobBuilder  # noqa: F401
from dlrover_amd.unified.master import PrimeMaster  # noqa: F401
